"""Generalized-to-standard eigenproblem transform (HEGST, itype=1, Lower).

Counterpart of ``eigensolver/gen_to_std/impl.h`` (local 222-285, distributed
287-500): A <- inv(L) A inv(L)^H in place, given the Cholesky factor L of B.

Per-k skeleton (the reference's reshuffled form, which defers the large
trailing TRSM: "only the tasks involving the k-th panel of L are executed" —
``impl.h:360-364``):

  1. finalize row k of every PREVIOUS panel j<k (left-solve with L[k,k]) and
     propagate downward with L[:,k]            (deferred forward substitution)
  2. hegst of the diagonal tile
  3. panel k partial update: right-solve with inv(L[k,k])^H and the first
     -1/2 L[:,k] A[k,k] Hermitian correction
  4. trailing HER2K-shaped update  A[i,j] -= A[i,k] L[j,k]^H + L[i,k] A[j,k]^H
  5. second -1/2 L[:,k] A[k,k] correction

All tile products are fused-GEMM launches on GPU; panels are contiguous RCCL
broadcasts (direct + two-hop transposed).
"""

from __future__ import annotations

from typing import Optional

import torch

from ..types import UpLo, Op, is_complex
from ..matrix.matrix import Matrix
from ..matrix.panel import Panel
from ..comm.grid import CommGrid
from ..comm import collectives as coll
from ..ops import tile_ops as ops
from . import _panels as pan
from .triangular import _trivial_grid
from .multiplication import _herm_full


def _opc(dtype) -> Op:
    return Op.ConjTrans if is_complex(dtype) else Op.Trans


def _hegst_diag_tile(a: torch.Tensor, l: torch.Tensor) -> None:
    """A_kk <- inv(L_kk) herm(A_kk) inv(L_kk)^H, written as a full Hermitian tile."""
    linv = ops.tri_inverse_full(l, lower=True)
    ah = _herm_full(a, lower=True)
    if a.is_cuda:
        nb = a.shape[0]
        tmp = torch.empty_like(a)
        opc = _opc(a.dtype)
        ops.gemm_items(tmp, linv, ah, [(0, 0, 0)], nb, Op.NoTrans, Op.NoTrans, 1.0, 0.0)
        ops.gemm_items(a, tmp, linv, [(0, 0, 0)], nb, Op.NoTrans, opc, 1.0, 0.0)
    else:
        a.copy_(linv @ ah @ linv.mH)


def generalized_to_standard(uplo: UpLo, mat_a: Matrix, mat_l: Matrix,
                            grid: Optional[CommGrid] = None) -> None:
    """In-place HEGST (itype=1): A <- inv(L) A inv(L)^H (Lower) or
    A <- inv(U)^H A inv(U) (Upper storage, native on the local path)."""
    if uplo == UpLo.Upper:
        g = grid if grid is not None else mat_a.grid
        if g is None or not g.distributed:
            _hegst_local_upper(mat_a, mat_l)   # native, no transposes
            return
        # distributed Upper: storage-transpose reduction (future work)
        from ._uplo import transpose_storage
        transpose_storage(mat_a)
        transpose_storage(mat_l)
        generalized_to_standard(UpLo.Lower, mat_a, mat_l, grid)
        transpose_storage(mat_a)
        transpose_storage(mat_l)
        return
    da, dl = mat_a.dist, mat_l.dist
    assert da.m == da.n and da.mb == da.nb
    assert (dl.m, dl.n, dl.mb, dl.nb) == (da.m, da.n, da.mb, da.nb)
    g = _trivial_grid(grid if grid is not None else mat_a.grid)
    d = da
    nt = d.nr_tiles[0]
    nb = d.nb
    dev, dt = mat_a.device, mat_a.dtype
    opc = _opc(dt)
    lr, lc = d.local_nr_tiles

    colpL = Panel(Panel.COL, d, dt, dev)
    rowpL = Panel(Panel.ROW, d, dt, dev)
    colpA = Panel(Panel.COL, d, dt, dev)
    rowpA = Panel(Panel.ROW, d, dt, dev)
    rowpAp = Panel(Panel.ROW, d, dt, dev)   # finalized rows of previous panels
    l_ws = torch.empty((nb, nb), dtype=dt, device=dev)
    a_ws = torch.empty((nb, nb), dtype=dt, device=dev)

    # hoist the diagonal-block inversions (L is read-only here; each
    # tri_inverse_full is a multi-launch recursion off the per-k critical path)
    lkk_tiles = []
    for k in range(nt):
        Lkk = pan.bcast_diag_to_all(mat_l, g, k, l_ws)
        lkk_tiles.append(Lkk.clone())
    linvs = ops.tri_inverse_full_many(lkk_tiles, lower=True)

    for k in range(nt):
        kr, kc = d.rank_of_tile((k, k))
        li0k = d.next_local_tile_row(k)
        li0 = d.next_local_tile_row(k + 1)
        lj_prev = d.next_local_tile_col(k)
        lj0 = d.next_local_tile_col(k + 1)

        # L panels of column k (rows >= k), direct + transposed
        pan.bcast_col_panel(mat_l, g, k, li0k, lr, colpL)
        pan.transpose_col_to_row(d, g, colpL, rowpL, lj0, lc)
        Lkk = pan.bcast_diag_to_all(mat_l, g, k, l_ws)

        # 1) deferred forward substitution: finalize row k of panels j < k
        if k > 0:
            if d.rank_row == kr:
                linv = linvs[k]
                if dev.type == "cuda":
                    items = [(rowpAp.offset(lj), 0,
                              mat_a.tile_offset((k, d.global_tile_of_local((0, lj))[1])))
                             for lj in range(lj_prev)]
                    ops.gemm_items(rowpAp.storage, linv, mat_a.storage, items, nb,
                                   Op.NoTrans, Op.NoTrans, 1.0, 0.0, uniform=True)
                    lrk = d.local_tile_of_global((k, d.global_tile_of_local((0, 0))[1]))[0]
                    mat_a.storage[lrk, :lj_prev].copy_(rowpAp.storage[:lj_prev])
                else:
                    for lj in range(lj_prev):
                        j = d.global_tile_of_local((0, lj))[1]
                        t = mat_a.tile((k, j))
                        r = rowpAp.slot(lj)
                        torch.matmul(linv, t, out=r)
                        t.copy_(r)
            if g.col_group is not None and lj_prev > 0:
                coll.broadcast(rowpAp.range_view(0, lj_prev),
                               g.global_rank_of_col_member(kr), g.col_group)
            items = [(mat_a.local_tile_offset(li, lj), colpL.offset(li), rowpAp.offset(lj))
                     for li in range(li0, lr) for lj in range(lj_prev)]
            ops.gemm_items(mat_a.storage, colpL.storage, rowpAp.storage, items, nb,
                           Op.NoTrans, Op.NoTrans, -1.0, 1.0, uniform=True)

        # 2) diagonal tile transform
        if (d.rank_row, d.rank_col) == (kr, kc):
            _hegst_diag_tile(mat_a.tile((k, k)), mat_l.tile((k, k)))

        # 3) panel k partial update (ranks in the owning column)
        Akk = pan.bcast_diag_to_col(mat_a, g, k, a_ws)
        if d.rank_col == kc and li0 < lr:
            linv = linvs[k]
            offs = [mat_a.tile_offset((d.global_tile_of_local((li, 0))[0], k))
                    for li in range(li0, lr)]
            loffs = [mat_l.tile_offset((d.global_tile_of_local((li, 0))[0], k))
                     for li in range(li0, lr)]
            # A[i,k] = A[i,k] @ inv(L_kk)^H
            ops.gemm_items(mat_a.storage, mat_a.storage, linv,
                           [(o, o, 0) for o in offs], nb, Op.NoTrans, opc, 1.0, 0.0,
                           inplace=True)
            # A[i,k] -= 1/2 L[i,k] herm(A_kk)
            hA = _herm_full(Akk, lower=True)
            ops.gemm_items(mat_a.storage, mat_l.storage, hA,
                           [(o, lo, 0) for o, lo in zip(offs, loffs)], nb,
                           Op.NoTrans, Op.NoTrans, -0.5, 1.0, uniform=True)
        # broadcast the updated A panel (direct + transposed)
        pan.bcast_col_panel(mat_a, g, k, li0, lr, colpA)
        pan.transpose_col_to_row(d, g, colpA, rowpA, lj0, lc)

        # 4) trailing HER2K-shaped update: A[i,j] -= A[i,k] L[j,k]^H + L[i,k] A[j,k]^H
        items = []
        for lj in range(lj0, lc):
            j = d.global_tile_of_local((0, lj))[1]
            for li in range(d.next_local_tile_row(j), lr):
                items.append((mat_a.local_tile_offset(li, lj), li, lj))
        trip_a = [(c, colpA.offset(li), rowpL.offset(lj)) for c, li, lj in items]
        trip_l = [(c, colpL.offset(li), rowpA.offset(lj)) for c, li, lj in items]
        ops.gemm_items(mat_a.storage, colpA.storage, rowpL.storage, trip_a, nb,
                       Op.NoTrans, opc, -1.0, 1.0, uniform=True)
        ops.gemm_items(mat_a.storage, colpL.storage, rowpA.storage, trip_l, nb,
                       Op.NoTrans, opc, -1.0, 1.0, uniform=True)

        # 5) second -1/2 L[:,k] A[k,k] correction
        if d.rank_col == kc and li0 < lr:
            hA = _herm_full(Akk, lower=True)
            offs = [mat_a.tile_offset((d.global_tile_of_local((li, 0))[0], k))
                    for li in range(li0, lr)]
            loffs = [mat_l.tile_offset((d.global_tile_of_local((li, 0))[0], k))
                     for li in range(li0, lr)]
            ops.gemm_items(mat_a.storage, mat_l.storage, hA,
                           [(o, lo, 0) for o, lo in zip(offs, loffs)], nb,
                           Op.NoTrans, Op.NoTrans, -0.5, 1.0, uniform=True)


def _herm_full_u(t: torch.Tensor) -> torch.Tensor:
    """Full Hermitian tile from UPPER storage."""
    return torch.triu(t) + torch.triu(t, 1).mH


def _hegst_local_upper(mat_a: Matrix, mat_u: Matrix) -> None:
    """Native Upper HEGST (itype=1, local): A <- U^{-H} A U^{-1} on
    upper-stored tiles, the LAPACK zhegst Upper blocked order (reference
    counterpart ``eigensolver/gen_to_std/impl.h`` call_U; the Lower path
    uses the deferred-TRSM reshuffle — Upper is correctness-grade per-tile
    ops, no storage transposes).

    Per tile step k (B12 = U[k, k+1:], A12 = A[k, k+1:]):
      1. A_kk <- Ukk^{-H} herm(A_kk) Ukk^{-1}
      2. A12 <- Ukk^{-H} A12
      3. A12 -= 1/2 herm(A_kk_new) B12
      4. A22 -= A12^H B12 + B12^H A12    (upper tiles only)
      5. A12 -= 1/2 herm(A_kk_new) B12
      6. A12 <- A12 U22^{-1}             (forward column sweep over U22)
    """
    da, du = mat_a.dist, mat_u.dist
    assert da.m == da.n and da.mb == da.nb
    assert (du.m, du.n, du.mb, du.nb) == (da.m, da.n, da.mb, da.nb)
    nt = da.nr_tiles[0]
    sta, stu = mat_a.storage, mat_u.storage

    for k in range(nt):
        Ukk = torch.triu(stu[k, k])
        # 1) diagonal transform (solve twice against the upper factor)
        Ah = _herm_full_u(sta[k, k])
        T1 = torch.linalg.solve_triangular(Ukk.mH, Ah, upper=False)
        sta[k, k] = torch.linalg.solve_triangular(
            Ukk.mH, T1.mH, upper=False).mH
        if k + 1 >= nt:
            break
        Akk_h = _herm_full_u(sta[k, k])
        # 2) A12 <- Ukk^{-H} A12
        for j in range(k + 1, nt):
            sta[k, j] = torch.linalg.solve_triangular(Ukk.mH, sta[k, j],
                                                      upper=False)
        # 3) A12 -= 1/2 Akk B12
        for j in range(k + 1, nt):
            sta[k, j] -= 0.5 * (Akk_h @ stu[k, j])
        # 4) A22 -= A12^H B12 + B12^H A12 (upper triangle of tiles)
        for i in range(k + 1, nt):
            for j in range(i, nt):
                sta[i, j] -= sta[k, i].mH @ stu[k, j] + \
                    stu[k, i].mH @ sta[k, j]
        # 5) second half-update
        for j in range(k + 1, nt):
            sta[k, j] -= 0.5 * (Akk_h @ stu[k, j])
        # 6) A12 <- A12 U22^{-1}: forward sweep over the trailing columns
        for j in range(k + 1, nt):
            acc = sta[k, j].clone()
            for i in range(k + 1, j):
                acc -= sta[k, i] @ stu[i, j]
            sta[k, j] = torch.linalg.solve_triangular(
                torch.triu(stu[j, j]).mH, acc.mH, upper=False).mH
        del Akk_h
