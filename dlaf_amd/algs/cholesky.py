"""Tiled Cholesky factorization (POTRF), local and distributed.

Counterpart of the reference's ``factorization/cholesky/impl.h`` (local
151-189, distributed 192-313): right-looking algorithm

    for k:  L_kk = potrf(A_kk)
            L_ik = A_ik * L_kk^-H              (panel TRSM, i > k)
            A_ij -= L_ik * L_jk^H              (trailing HERK/GEMM, i >= j > k)

MI355X-native differences from the reference:
* per-tile task flood -> FUSED kernels: the whole panel solve is a handful of
  launches and the whole trailing update is ONE launch per k (descriptor list
  over the rank's local tiles);
* panel TRSM runs against diagonal-block inverses computed during potrf_tile
  (TRSM-as-GEMM, csrc/factor.hip);
* MPI Ibcast chains -> RCCL broadcasts on the row/col process groups: the
  column panel is ONE contiguous broadcast along the row direction, the
  transposed row panel is per-tile broadcasts along the column direction
  (reference ``broadcast_panel.h:78-189`` semantics).

Only UpLo.Lower is implemented (the reference's miniapps and C API default);
Upper can be added by symmetry.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..types import UpLo, Op, is_complex
from ..matrix.matrix import Matrix
from ..matrix.panel import Panel
from ..comm.grid import CommGrid
from ..comm import collectives as coll
from ..ops import tile_ops as ops


def _op_conj(dtype) -> Op:
    return Op.ConjTrans if is_complex(dtype) else Op.Trans


def _cpu_trsm_tile(diag: torch.Tensor, tile: torch.Tensor) -> None:
    # X * diag^H = tile  ->  solve_triangular(diag^H upper, left=False)
    X = torch.linalg.solve_triangular(diag.mH, tile, upper=True, left=False)
    tile.copy_(X)


def _compute_dinv(diag: torch.Tensor) -> torch.Tensor:
    """Diagonal-block inverses of a (lower-triangular) padded tile (GPU)."""
    nb = diag.shape[0]
    dinv = ops.dinv_workspace(nb, diag.dtype, diag.device)
    bsz = dinv.shape[-1]
    ext = ops.get_ext()
    for d in range((nb + bsz - 1) // bsz):
        c0 = d * bsz
        bs = min(bsz, nb - c0)
        ext.trtri_lower(diag[c0:, c0:], dinv[d], bs, diag.stride(0), bsz, False)
    return dinv


def _trailing_update_local(mat: Matrix, k: int) -> None:
    d = mat.dist
    nt = d.nr_tiles[0]
    nb = d.nb
    opc = _op_conj(mat.dtype)
    if mat.device.type == "cuda":
        c_offs, a_offs, b_offs = [], [], []
        for j in range(k + 1, nt):
            for i in range(j, nt):
                c_offs.append(mat.tile_offset((i, j)))
                a_offs.append(mat.tile_offset((i, k)))
                b_offs.append(mat.tile_offset((j, k)))
        if not c_offs:
            return
        descs = ops.make_descs(c_offs, a_offs, b_offs)
        st = mat.storage
        ops.gemm_fused(st, st, st, descs, nb, nb, nb, nb, nb, nb,
                       Op.NoTrans, opc, -1.0, 1.0)
    else:
        for j in range(k + 1, nt):
            for i in range(j, nt):
                ops.gemm_tile(mat.tile((i, j)), mat.tile((i, k)), mat.tile((j, k)),
                              Op.NoTrans, opc, -1.0, 1.0)


def _cholesky_local(mat: Matrix) -> None:
    d = mat.dist
    nt = d.nr_tiles[0]
    nb = d.nb
    gpu = mat.device.type == "cuda"
    dinv = ops.dinv_workspace(nb, mat.dtype, mat.device) if gpu else None
    for k in range(nt):
        diag = mat.tile((k, k))
        ops.potrf_tile(diag, dinv)
        rows = list(range(k + 1, nt))
        if rows:
            if gpu:
                offs = [mat.tile_offset((i, k)) for i in rows]
                ops.trsm_panel_right_lowerH(mat.storage, offs, diag, dinv, nb, nb, nb)
            else:
                for i in rows:
                    _cpu_trsm_tile(diag, mat.tile((i, k)))
        _trailing_update_local(mat, k)


def _cholesky_dist(mat: Matrix, grid: CommGrid) -> None:
    d = mat.dist
    nt = d.nr_tiles[0]
    nb = d.nb
    gpu = mat.device.type == "cuda"
    opc = _op_conj(mat.dtype)

    col_panel = Panel(Panel.COL, d, mat.dtype, mat.device)  # L[i,k] by local row
    row_panel = Panel(Panel.ROW, d, mat.dtype, mat.device)  # L[j,k] by local col
    diag_ws = torch.zeros((nb, nb), dtype=mat.dtype, device=mat.device)

    for k in range(nt):
        kr, kc = d.rank_of_tile((k, k))
        on_panel_col = d.rank_col == kc
        li0 = d.next_local_tile_row(k + 1)  # first local row tile strictly below k
        lr = d.local_nr_tiles[0]

        diag = None
        dinv = None
        if on_panel_col:
            # --- diagonal factor + broadcast down the column ---
            if d.rank_row == kr:
                diag = mat.tile((k, k))
                dinv = ops.potrf_tile(diag, None)
            else:
                diag = diag_ws
            if grid.col_group is not None:
                coll.broadcast(diag, grid.global_rank_of_col_member(kr), grid.col_group)
            if gpu and dinv is None:
                dinv = _compute_dinv(diag)
            # --- panel TRSM on my local rows below k ---
            rows = [d.global_tile_of_local((li, 0))[0] for li in range(li0, lr)]
            if rows:
                if gpu:
                    offs = [mat.tile_offset((i, k)) for i in rows]
                    ops.trsm_panel_right_lowerH(mat.storage, offs, diag, dinv, nb, nb, nb)
                else:
                    for i in rows:
                        _cpu_trsm_tile(diag, mat.tile((i, k)))
            # stage panel into the contiguous broadcast workspace
            for li in range(li0, lr):
                i = d.global_tile_of_local((li, 0))[0]
                col_panel.slot(li).copy_(mat.tile((i, k)))

        # --- column panel: one contiguous broadcast along the row direction ---
        if grid.row_group is not None and lr - li0 > 0:
            coll.broadcast(
                col_panel.range_view(li0, lr),
                grid.global_rank_of_row_member(kc),
                grid.row_group,
            )

        # --- transposed panel: per-tile broadcasts along the column direction ---
        lj0 = d.next_local_tile_col(k + 1)
        lc = d.local_nr_tiles[1]
        for lj in range(lj0, lc):
            j = d.global_tile_of_local((0, lj))[1]
            jr = d.rank_of_tile_row(j)
            if d.rank_row == jr:
                lji = d.next_local_tile_row(j)  # local row index of global row j
                row_panel.slot(lj).copy_(col_panel.slot(lji))
            if grid.col_group is not None:
                coll.broadcast(
                    row_panel.slot(lj), grid.global_rank_of_col_member(jr), grid.col_group
                )

        # --- trailing update on local tiles (global i >= j > k) ---
        if gpu:
            c_offs, a_offs, b_offs = [], [], []
            for lj in range(lj0, lc):
                j = d.global_tile_of_local((0, lj))[1]
                for li in range(d.next_local_tile_row(j), lr):
                    c_offs.append(mat.local_tile_offset(li, lj))
                    a_offs.append(col_panel.offset(li))
                    b_offs.append(row_panel.offset(lj))
            if c_offs:
                descs = ops.make_descs(c_offs, a_offs, b_offs)
                ops.gemm_fused(
                    mat.storage, col_panel.storage, row_panel.storage, descs,
                    nb, nb, nb, nb, nb, nb, Op.NoTrans, opc, -1.0, 1.0,
                )
        else:
            for lj in range(lj0, lc):
                j = d.global_tile_of_local((0, lj))[1]
                for li in range(d.next_local_tile_row(j), lr):
                    ops.gemm_tile(
                        mat.local_tile((li, lj)), col_panel.slot(li), row_panel.slot(lj),
                        Op.NoTrans, opc, -1.0, 1.0,
                    )


def cholesky_factorization(uplo: UpLo, mat: Matrix, grid: Optional[CommGrid] = None) -> None:
    """In-place Cholesky factorization A = L L^H (Lower) of a tiled matrix."""
    assert uplo == UpLo.Lower, "only Lower implemented (as in the reference miniapps)"
    d = mat.dist
    assert d.m == d.n and d.mb == d.nb, "square matrix with square tiles required"
    g = grid if grid is not None else mat.grid
    if g is None or not g.distributed:
        _cholesky_local(mat)
    else:
        _cholesky_dist(mat, g)
