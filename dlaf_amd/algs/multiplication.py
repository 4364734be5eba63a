"""Hermitian and general tiled multiplication.

Counterpart of ``multiplication/hermitian/impl.h:69-213`` (HEMM: Left/Right x
Lower/Upper) and ``multiplication/general/impl.h:35-153`` (``GeneralSub::callNN``
— the D&C eigensolver's workhorse GEMM; like the reference, only NoTrans x
NoTrans is supported distributed, all ops locally).

Per-k skeleton (Left HEMM): broadcast B row-panel k (col direction), broadcast
the A column/row panels (direct for the stored triangle, two-hop transposed +
conjugate for the mirrored triangle), one fused-GEMM launch per part across
all local C tiles, Hermitian-completed diagonal block product.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..types import Side, UpLo, Op, is_complex
from ..matrix.matrix import Matrix
from ..matrix.panel import Panel
from ..comm.grid import CommGrid
from ..ops import tile_ops as ops
from . import _panels as pan
from .triangular import (
    _bcast_B_col,
    _bcast_B_row,
    _trivial_grid,
    _update_tiles,
)


def _herm_full(tile: torch.Tensor, lower: bool) -> torch.Tensor:
    """Hermitian completion of a diagonal tile stored in one triangle."""
    if lower:
        low = torch.tril(tile)
        return low + torch.tril(tile, -1).mH
    up = torch.triu(tile)
    return up + torch.triu(tile, 1).mH


def _opc(dtype) -> Op:
    return Op.ConjTrans if is_complex(dtype) else Op.Trans


def hermitian_multiplication(side: Side, uplo: UpLo, alpha, A: Matrix, B: Matrix,
                             beta, C: Matrix, grid: Optional[CommGrid] = None) -> None:
    """C <- alpha A B + beta C (Left) or alpha B A + beta C (Right), A Hermitian.

    Only the ``uplo`` triangle of A is referenced. Reference:
    ``multiplication/hermitian/impl.h:69-213``.
    """
    da, db, dc = A.dist, B.dist, C.dist
    assert da.m == da.n and da.mb == da.nb
    assert db.mb == db.nb == da.nb and dc.mb == dc.nb == da.nb
    assert (db.m, db.n) == (dc.m, dc.n)
    assert da.m == (dc.m if side == Side.Left else dc.n)
    g = _trivial_grid(grid if grid is not None else C.grid)
    lower = uplo == UpLo.Lower
    nt = da.nr_tiles[0]
    nb = da.nb
    dev, dt = C.device, C.dtype
    opc = _opc(dt)

    if beta != 1:
        C.storage.mul_(beta)
    diag_ws = torch.empty((nb, nb), dtype=dt, device=dev)
    lrC, lcC = dc.local_nr_tiles

    if side == Side.Left:
        rowpB = Panel(Panel.ROW, db, dt, dev)
        colpA = Panel(Panel.COL, da, dt, dev)
        rowpA = Panel(Panel.ROW, da, dt, dev)
        for k in range(nt):
            _bcast_B_row(B, g, k, rowpB, fill=True)
            # stored-triangle part (direct col panel of A)
            li0d, li1d = ((da.next_local_tile_row(k + 1), da.local_nr_tiles[0])
                          if lower else (0, da.next_local_tile_row(k)))
            pan.bcast_col_panel(A, g, k, li0d, li1d, colpA)
            items = [(li, lj) for li in range(li0d, li1d) for lj in range(lcC)]
            _update_tiles(C, items, colpA, rowpB, Op.NoTrans, Op.NoTrans, alpha, 1.0)
            # mirrored-triangle part: A_herm[i,k] = A[k,i]^H (two-hop transpose)
            li0t, li1t = ((0, da.next_local_tile_row(k)) if lower
                          else (da.next_local_tile_row(k + 1), da.local_nr_tiles[0]))
            lj0, lj1 = ((0, da.next_local_tile_col(k)) if lower
                        else (da.next_local_tile_col(k + 1), da.local_nr_tiles[1]))
            pan.bcast_row_panel(A, g, k, lj0, lj1, rowpA)
            pan.transpose_row_to_col(da, g, rowpA, colpA, li0t, li1t)
            items = [(li, lj) for li in range(li0t, li1t) for lj in range(lcC)]
            _update_tiles(C, items, colpA, rowpB, opc, Op.NoTrans, alpha, 1.0)
            # Hermitian diagonal block
            kr = da.rank_of_tile_row(k)
            dtile = pan.bcast_diag_to_row(A, g, k, diag_ws)
            if dc.rank_row == kr:
                hd = _herm_full(dtile, lower)
                c = [C.tile_offset((k, dc.global_tile_of_local((0, lj))[1])) for lj in range(lcC)]
                b = [rowpB.offset(lj) for lj in range(lcC)]
                if lcC:
                    if dev.type == "cuda":
                        ops.gemm_fused(C.storage, hd, rowpB.storage, ops.make_descs(c, [0] * len(c), b),
                                       nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.NoTrans, alpha, 1.0,
                                       uniform=True)
                    else:
                        for lj in range(lcC):
                            j = dc.global_tile_of_local((0, lj))[1]
                            C.tile((k, j)).add_(hd @ rowpB.slot(lj), alpha=alpha)
    else:
        colpB = Panel(Panel.COL, db, dt, dev)
        rowpA = Panel(Panel.ROW, da, dt, dev)
        colpA = Panel(Panel.COL, da, dt, dev)
        for k in range(nt):
            _bcast_B_col(B, g, k, colpB, fill=True)
            # stored-triangle part: A_herm[k,j] = A[k,j] direct (Lower: j<k)
            lj0d, lj1d = ((0, da.next_local_tile_col(k)) if lower
                          else (da.next_local_tile_col(k + 1), da.local_nr_tiles[1]))
            pan.bcast_row_panel(A, g, k, lj0d, lj1d, rowpA)
            items = [(li, lj) for li in range(lrC) for lj in range(lj0d, lj1d)]
            _update_tiles(C, items, colpB, rowpA, Op.NoTrans, Op.NoTrans, alpha, 1.0)
            # mirrored part: A_herm[k,j] = A[j,k]^H (Lower: j>k)
            lj0t, lj1t = ((da.next_local_tile_col(k + 1), da.local_nr_tiles[1]) if lower
                          else (0, da.next_local_tile_col(k)))
            li0, li1 = ((da.next_local_tile_row(k + 1), da.local_nr_tiles[0]) if lower
                        else (0, da.next_local_tile_row(k)))
            pan.bcast_col_panel(A, g, k, li0, li1, colpA)
            pan.transpose_col_to_row(da, g, colpA, rowpA, lj0t, lj1t)
            items = [(li, lj) for li in range(lrC) for lj in range(lj0t, lj1t)]
            _update_tiles(C, items, colpB, rowpA, Op.NoTrans, opc, alpha, 1.0)
            # Hermitian diagonal block
            kc = da.rank_of_tile_col(k)
            dtile = pan.bcast_diag_to_col(A, g, k, diag_ws)
            if dc.rank_col == kc:
                hd = _herm_full(dtile, lower)
                c = [C.tile_offset((dc.global_tile_of_local((li, 0))[0], k)) for li in range(lrC)]
                a = [colpB.offset(li) for li in range(lrC)]
                if lrC:
                    if dev.type == "cuda":
                        ops.gemm_fused(C.storage, colpB.storage, hd, ops.make_descs(c, a, [0] * len(c)),
                                       nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.NoTrans, alpha, 1.0,
                                       uniform=True)
                    else:
                        for li in range(lrC):
                            i = dc.global_tile_of_local((li, 0))[0]
                            C.tile((i, k)).add_(colpB.slot(li) @ hd, alpha=alpha)


def general_multiplication(opA: Op, opB: Op, alpha, A: Matrix, B: Matrix,
                           beta, C: Matrix, grid: Optional[CommGrid] = None) -> None:
    """C <- alpha op(A) op(B) + beta C over tiled matrices.

    Distributed: NoTrans x NoTrans only (as the reference,
    ``multiplication/general.h:61-64``); local: all op combinations.
    """
    g = _trivial_grid(grid if grid is not None else C.grid)
    da, db, dc = A.dist, B.dist, C.dist
    nb = dc.nb
    assert da.mb == da.nb == nb and db.mb == db.nb == nb and dc.mb == nb

    if not g.distributed:
        _general_local(opA, opB, alpha, A, B, beta, C)
        return
    assert opA is Op.NoTrans and opB is Op.NoTrans, \
        "distributed general_multiplication supports NoTrans x NoTrans only (as the reference)"
    assert da.m == dc.m and db.n == dc.n and da.n == db.m
    if beta != 1:
        C.storage.mul_(beta)
    dev, dt = C.device, C.dtype
    lrC, lcC = dc.local_nr_tiles
    colpA = Panel(Panel.COL, dc, dt, dev)
    rowpB = Panel(Panel.ROW, dc, dt, dev)
    ntk = da.nr_tiles[1]
    items = [(li, lj) for li in range(lrC) for lj in range(lcC)]
    for k in range(ntk):
        pan.bcast_col_panel(A, g, k, 0, lrC, colpA)
        pan.bcast_row_panel(B, g, k, 0, lcC, rowpB)
        _update_tiles(C, items, colpA, rowpB, Op.NoTrans, Op.NoTrans, alpha, 1.0)


def _general_local(opA: Op, opB: Op, alpha, A: Matrix, B: Matrix, beta, C: Matrix) -> None:
    ntr, ntc = C.dist.nr_tiles
    ntk = A.dist.nr_tiles[1] if opA is Op.NoTrans else A.dist.nr_tiles[0]
    nb = C.dist.nb

    def a_tile(i, k):
        return A.tile((i, k)) if opA is Op.NoTrans else A.tile((k, i))

    def b_tile(k, j):
        return B.tile((k, j)) if opB is Op.NoTrans else B.tile((j, k))

    if C.device.type == "cuda" and ntk > 0:
        # one fused launch: per-C-tile K-loop over ntk operand tiles
        ts = nb * nb
        lcA = A.dist.local_nr_tiles[1]
        lcB = B.dist.local_nr_tiles[1]
        lcC = C.dist.local_nr_tiles[1]
        a_ks = ts if opA is Op.NoTrans else lcA * ts
        b_ks = lcB * ts if opB is Op.NoTrans else ts
        c_offs, a_offs, b_offs = [], [], []
        for i in range(ntr):
            for j in range(ntc):
                c_offs.append((i * lcC + j) * ts)
                a_offs.append((i * lcA) * ts if opA is Op.NoTrans else i * ts)
                b_offs.append(j * ts if opB is Op.NoTrans else (j * lcB) * ts)
        descs = ops.make_descs(c_offs, a_offs, b_offs, ktiles=ntk, a_kstride=a_ks, b_kstride=b_ks)
        ops.gemm_fused(C.storage, A.storage, B.storage, descs,
                       nb, nb, nb, nb, nb, nb, opA, opB, alpha, beta)
        return
    for i in range(ntr):
        for j in range(ntc):
            ct = C.tile((i, j))
            if beta != 1:
                ct.mul_(beta)
            for k in range(ntk):
                ops.gemm_tile(ct, a_tile(i, k), b_tile(k, j), opA, opB, alpha, 1.0)
