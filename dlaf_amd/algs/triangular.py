"""Distributed triangular solve (TRSM) and multiplication (TRMM).

Counterpart of the reference's ``solver/triangular/impl.h:238-1205`` and
``multiplication/triangular/impl.h:203-726``: all 8 {side, uplo, op} cases,
local and distributed, with the reference's per-k skeleton

    TRSM:  bcast diag -> solve B row/col k -> bcast solved panel ->
           bcast A panel -> trailing GEMM update            (impl.h:482-575 LLN)
    TRMM:  bcast B panel (pre-multiply value) -> trailing GEMM accumulation ->
           diag triangular multiply

MI355X-native design: per-tile triangular solves become GEMMs against a
full-tile inverse produced by the native ``trtri_lower`` CDNA4 kernel
(TRSM-as-GEMM); every trailing update across the rank's local tiles is ONE
fused-GEMM launch (``csrc/gemm_tiles.hip``); panel broadcasts are contiguous
RCCL collectives (``_panels.py``). CPU path: per-tile torch.linalg — the
reference backend the GPU path is tested against.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..types import Side, UpLo, Op, Diag
from ..matrix.matrix import Matrix
from ..matrix.panel import Panel
from ..comm.grid import CommGrid
from ..comm import collectives as coll
from ..ops import tile_ops as ops
from . import _panels as pan


def _t(x: torch.Tensor, op: Op) -> torch.Tensor:
    if op is Op.NoTrans:
        return x
    if op is Op.Trans:
        return x.mT
    return x.mH


def _trivial_grid(grid: Optional[CommGrid]) -> CommGrid:
    return grid if grid is not None else CommGrid(1, 1)


# ---------------------------------------------------------------------------
# fused / looped building blocks (GPU: one kernel launch; CPU: per-tile torch)
# ---------------------------------------------------------------------------

def _update_tiles(mat: Matrix, items: List[Tuple[int, int]], colp: Panel, rowp: Panel,
                  opA: Op, opB: Op, alpha, beta) -> None:
    """mat[li,lj] = alpha * op(colp[li]) @ op(rowp[lj]) + beta * mat[li,lj]."""
    if not items:
        return
    nb = mat.dist.nb
    if mat.device.type == "cuda":
        c = [mat.local_tile_offset(li, lj) for li, lj in items]
        a = [colp.offset(li) for li, _ in items]
        b = [rowp.offset(lj) for _, lj in items]
        ops.gemm_fused(mat.storage, colp.storage, rowp.storage, ops.make_descs(c, a, b),
                       nb, nb, nb, nb, nb, nb, opA, opB, alpha, beta, uniform=True)
    else:
        for li, lj in items:
            ops.gemm_tile(mat.local_tile((li, lj)), colp.slot(li), rowp.slot(lj),
                          opA, opB, alpha, beta)


def _left_mul_row(mat: Matrix, k: int, M: torch.Tensor, opM: Op, rowp: Panel,
                  lj0: int, lj1: int) -> None:
    """B[k, j] <- op(M) @ B[k, j] for local cols [lj0, lj1), via rowp staging."""
    if lj1 <= lj0:
        return
    d = mat.dist
    nb = d.nb
    if mat.device.type == "cuda":
        c = [rowp.offset(lj) for lj in range(lj0, lj1)]
        a = [0] * len(c)
        b = [mat.tile_offset((k, d.global_tile_of_local((0, lj))[1])) for lj in range(lj0, lj1)]
        ops.gemm_fused(rowp.storage, M, mat.storage, ops.make_descs(c, a, b),
                       nb, nb, nb, M.stride(0), nb, nb, opM, Op.NoTrans, 1.0, 0.0,
                       uniform=True)
        lrk = d.local_tile_of_global((k, d.global_tile_of_local((0, lj0))[1]))[0]
        mat.storage[lrk, lj0:lj1].copy_(rowp.storage[lj0:lj1])
    else:
        for lj in range(lj0, lj1):
            j = d.global_tile_of_local((0, lj))[1]
            t = mat.tile((k, j))
            r = rowp.slot(lj)
            torch.matmul(_t(M, opM), t, out=r)
            t.copy_(r)


def _right_mul_col_inplace(mat: Matrix, k: int, M: torch.Tensor, opM: Op,
                           li0: int, li1: int) -> None:
    """B[i, k] <- B[i, k] @ op(M) in place for local rows [li0, li1)."""
    if li1 <= li0:
        return
    d = mat.dist
    nb = d.nb
    if mat.device.type == "cuda":
        offs = [mat.tile_offset((d.global_tile_of_local((li, 0))[0], k)) for li in range(li0, li1)]
        ops.gemm_fused(mat.storage, mat.storage, M, ops.make_descs(offs, offs, [0] * len(offs)),
                       nb, nb, nb, nb, M.stride(0), nb, Op.NoTrans, opM, 1.0, 0.0, inplace=True)
    else:
        for li in range(li0, li1):
            i = d.global_tile_of_local((li, 0))[0]
            t = mat.tile((i, k))
            t.copy_(t @ _t(M, opM))


def _col_from_panel_mul(mat: Matrix, k: int, colp: Panel, M: torch.Tensor, opM: Op,
                        li0: int, li1: int) -> None:
    """B[i, k] <- colp[li] @ op(M) (TRMM right diag multiply; colp holds old B[:,k])."""
    if li1 <= li0:
        return
    d = mat.dist
    nb = d.nb
    if mat.device.type == "cuda":
        c = [mat.tile_offset((d.global_tile_of_local((li, 0))[0], k)) for li in range(li0, li1)]
        a = [colp.offset(li) for li in range(li0, li1)]
        ops.gemm_fused(mat.storage, colp.storage, M, ops.make_descs(c, a, [0] * len(c)),
                       nb, nb, nb, nb, M.stride(0), nb, Op.NoTrans, opM, 1.0, 0.0,
                       uniform=True)
    else:
        for li in range(li0, li1):
            i = d.global_tile_of_local((li, 0))[0]
            torch.matmul(colp.slot(li), _t(M, opM), out=mat.tile((i, k)))


def _row_from_panel_mul(mat: Matrix, k: int, rowp: Panel, M: torch.Tensor, opM: Op,
                        lj0: int, lj1: int) -> None:
    """B[k, j] <- op(M) @ rowp[lj] (TRMM left diag multiply; rowp holds old B[k,:])."""
    if lj1 <= lj0:
        return
    d = mat.dist
    nb = d.nb
    if mat.device.type == "cuda":
        c = [mat.tile_offset((k, d.global_tile_of_local((0, lj))[1])) for lj in range(lj0, lj1)]
        b = [rowp.offset(lj) for lj in range(lj0, lj1)]
        ops.gemm_fused(mat.storage, M, rowp.storage, ops.make_descs(c, [0] * len(c), b),
                       nb, nb, nb, M.stride(0), nb, nb, opM, Op.NoTrans, 1.0, 0.0,
                       uniform=True)
    else:
        for lj in range(lj0, lj1):
            j = d.global_tile_of_local((0, lj))[1]
            torch.matmul(_t(M, opM), rowp.slot(lj), out=mat.tile((k, j)))


# ---------------------------------------------------------------------------
# A-panel preparation (direct or transposed two-hop broadcast)
# ---------------------------------------------------------------------------

def _prep_left_A_panel(A: Matrix, grid: CommGrid, k: int, below: bool,
                       transposed: bool, colpA: Panel, rowpA: Panel) -> Tuple[int, int]:
    """Fill colpA (keyed by local tile-row) for the update row range of step k.

    Direct (transposed=False): colpA[li] = A[i, k].
    Transposed:                colpA[li] = A[k, i] (two-hop, reference
    ``broadcast_panel.h:78-189`` semantics).
    Returns the local row slot range [li0, li1).
    """
    da = A.dist
    lrA = da.local_nr_tiles[0]
    li0, li1 = (da.next_local_tile_row(k + 1), lrA) if below else (0, da.next_local_tile_row(k))
    if not transposed:
        pan.bcast_col_panel(A, grid, k, li0, li1, colpA)
    else:
        lcA = da.local_nr_tiles[1]
        lj0, lj1 = (da.next_local_tile_col(k + 1), lcA) if below else (0, da.next_local_tile_col(k))
        pan.bcast_row_panel(A, grid, k, lj0, lj1, rowpA)
        pan.transpose_row_to_col(da, grid, rowpA, colpA, li0, li1)
    return li0, li1


def _prep_right_A_panel(A: Matrix, grid: CommGrid, k: int, after: bool,
                        transposed: bool, rowpA: Panel, colpA: Panel) -> Tuple[int, int]:
    """Fill rowpA (keyed by local tile-col) for the update col range of step k.

    Direct: rowpA[lj] = A[k, j].   Transposed: rowpA[lj] = A[j, k].
    Returns the local col slot range [lj0, lj1).
    """
    da = A.dist
    lcA = da.local_nr_tiles[1]
    lj0, lj1 = (da.next_local_tile_col(k + 1), lcA) if after else (0, da.next_local_tile_col(k))
    if not transposed:
        pan.bcast_row_panel(A, grid, k, lj0, lj1, rowpA)
    else:
        lrA = da.local_nr_tiles[0]
        li0, li1 = (da.next_local_tile_row(k + 1), lrA) if after else (0, da.next_local_tile_row(k))
        pan.bcast_col_panel(A, grid, k, li0, li1, colpA)
        pan.transpose_col_to_row(da, grid, colpA, rowpA, lj0, lj1)
    return lj0, lj1


def _bcast_B_row(B: Matrix, grid: CommGrid, k: int, rowpB: Panel, fill: bool) -> Tuple[int, int]:
    """rowpB[lj] <- B[k, j] for ALL local cols, broadcast along the col direction."""
    d = B.dist
    lc = d.local_nr_tiles[1]
    kr = d.rank_of_tile_row(k)
    if fill and d.rank_row == kr:
        for lj in range(lc):
            j = d.global_tile_of_local((0, lj))[1]
            rowpB.slot(lj).copy_(B.tile((k, j)))
    if grid.col_group is not None and lc > 0:
        coll.broadcast(rowpB.range_view(0, lc), grid.global_rank_of_col_member(kr), grid.col_group)
    return 0, lc


def _bcast_B_col(B: Matrix, grid: CommGrid, k: int, colpB: Panel, fill: bool) -> Tuple[int, int]:
    """colpB[li] <- B[i, k] for ALL local rows, broadcast along the row direction."""
    d = B.dist
    lr = d.local_nr_tiles[0]
    kc = d.rank_of_tile_col(k)
    if fill and d.rank_col == kc:
        for li in range(lr):
            i = d.global_tile_of_local((li, 0))[0]
            colpB.slot(li).copy_(B.tile((i, k)))
    if grid.row_group is not None and lr > 0:
        coll.broadcast(colpB.range_view(0, lr), grid.global_rank_of_row_member(kc), grid.row_group)
    return 0, lr


def _check_args(side: Side, A: Matrix, B: Matrix) -> None:
    da, db = A.dist, B.dist
    assert da.m == da.n and da.mb == da.nb, "A must be square with square tiles"
    assert db.mb == db.nb and da.nb == db.nb, "equal square tile sizes required"
    if side == Side.Left:
        assert da.m == db.m, (da.m, db.m)
    else:
        assert da.m == db.n, (da.m, db.n)
    assert (da.grid_rows, da.grid_cols) == (db.grid_rows, db.grid_cols)


# ---------------------------------------------------------------------------
# TRSM
# ---------------------------------------------------------------------------

def _trsm_lln_local_gpu(diagv: Diag, A: Matrix, B: Matrix) -> None:
    """Local GPU fast path for Left-Lower-NoTrans (the BASELINE TRSM case):
    lookahead schedule mirroring Cholesky — the row-(k+1) solve runs on the
    high-priority stream overlapped with the bulk trailing update, descs are
    read straight off the matrix storage (no panel staging)."""
    from ..runtime import get_runtime
    da, db = A.dist, B.dist
    nt = da.nr_tiles[0]
    nb = da.nb
    ntc = db.nr_tiles[1]
    unit = diagv == Diag.Unit
    # precompute all diagonal-block inverses (off the critical path)
    invs = ops.tri_inverse_full_many((A.tile((k, k)) for k in range(nt)),
                                     lower=True, unit=unit)
    rt = get_runtime(B.device)
    sp, su = rt.hp_streams[0], rt.np_streams[0]
    cur = torch.cuda.current_stream(B.device)
    sp.wait_stream(cur)
    su.wait_stream(cur)
    rowbufs = [torch.empty((ntc, nb, nb), dtype=B.dtype, device=B.device)
               for _ in range(2)]
    ev_head = [None] * nt
    ev_tail = [None] * nt

    def row_offs(i):
        return [B.tile_offset((i, j)) for j in range(ntc)]

    for k in range(nt):
        rowbuf = rowbufs[k % 2]
        with torch.cuda.stream(sp):
            # row k's last pending updates: head(k-1) (column k-1 term) and
            # tail(k-2) (all earlier columns); ev_tail[k-2] also frees buffer
            # k%2 (last read by tail(k-2)).
            if k >= 1 and ev_head[k - 1] is not None:
                sp.wait_event(ev_head[k - 1])
            if k >= 2 and ev_tail[k - 2] is not None:
                sp.wait_event(ev_tail[k - 2])
            # solve row k: rowbuf = inv @ B[k, :], copy back
            offs = row_offs(k)
            c = [j * nb * nb for j in range(ntc)]
            ops.gemm_fused(rowbuf, invs[k], B.storage, ops.make_descs(c, [0] * ntc, offs),
                           nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.NoTrans, 1.0, 0.0,
                           uniform=True)
            # whole local row k is contiguous storage: ONE copy, not ntc
            B.storage[k, :ntc].copy_(rowbuf[:ntc])
            ev_s = torch.cuda.Event()
            ev_s.record(sp)
        # all accumulations into rows > k run on su (no cross-stream writes):
        # head (row k+1, small — unblocks solve(k+1)) first, then the bulk tail
        if k + 1 < nt:
            with torch.cuda.stream(su):
                su.wait_event(ev_s)
                ch = row_offs(k + 1)
                ak = [A.tile_offset((k + 1, k))] * ntc
                ops.gemm_fused(B.storage, A.storage, rowbuf, ops.make_descs(ch, ak, c),
                               nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.NoTrans, -1.0, 1.0,
                               uniform=True)
                eh = torch.cuda.Event()
                eh.record(su)
                ev_head[k] = eh
                if k + 2 < nt:
                    ct, at, bt = [], [], []
                    for i in range(k + 2, nt):
                        ct += row_offs(i)
                        at += [A.tile_offset((i, k))] * ntc
                        bt += [j * nb * nb for j in range(ntc)]
                    ops.gemm_fused(B.storage, A.storage, rowbuf, ops.make_descs(ct, at, bt),
                                   nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.NoTrans, -1.0, 1.0,
                                   uniform=True)
                    e = torch.cuda.Event()
                    e.record(su)
                    ev_tail[k] = e
    cur.wait_stream(sp)
    cur.wait_stream(su)


def triangular_solver(side: Side, uplo: UpLo, op: Op, diag: Diag, alpha,
                      A: Matrix, B: Matrix, grid: Optional[CommGrid] = None) -> None:
    """In-place solve op(A) X = alpha B (Left) or X op(A) = alpha B (Right).

    Reference: ``solver/triangular/impl.h`` (all 8 cases; distributed LLN at
    482-575). A is triangular (only the ``uplo`` triangle is referenced).
    """
    _check_args(side, A, B)
    g = _trivial_grid(grid if grid is not None else B.grid)
    unit = diag == Diag.Unit
    lower = uplo == UpLo.Lower
    notrans = op == Op.NoTrans
    if alpha != 1:
        B.storage.mul_(alpha)
    if (side == Side.Left and lower and notrans and not g.distributed
            and B.device.type == "cuda"):
        _trsm_lln_local_gpu(diag, A, B)
        return

    da, db = A.dist, B.dist
    nt = da.nr_tiles[0]
    nb = da.nb
    lrB, lcB = db.local_nr_tiles
    dev, dt = B.device, B.dtype
    diag_ws = torch.empty((nb, nb), dtype=dt, device=dev)

    if side == Side.Left:
        below = lower == notrans          # update rows beyond k
        transposed = not notrans          # A tiles come from row k
        order = range(nt) if below else range(nt - 1, -1, -1)
        colpA = Panel(Panel.COL, da, dt, dev)
        rowpA = Panel(Panel.ROW, da, dt, dev) if transposed else None
        rowpB = Panel(Panel.ROW, db, dt, dev)
        opA_upd = op if transposed else Op.NoTrans
        for k in order:
            kr = da.rank_of_tile_row(k)
            dtile = pan.bcast_diag_to_row(A, g, k, diag_ws)
            # solve B[k, :] on the owning rank row, stage into rowpB
            if db.rank_row == kr:
                inv = ops.tri_inverse_full(dtile, lower, unit)
                _left_mul_row(B, k, inv, op, rowpB, 0, lcB)
            _bcast_B_row(B, g, k, rowpB, fill=False)  # staged by the solve above
            li0, li1 = _prep_left_A_panel(A, g, k, below, transposed, colpA, rowpA)
            items = [(li, lj) for li in range(li0, li1) for lj in range(lcB)]
            _update_tiles(B, items, colpA, rowpB, opA_upd, Op.NoTrans, -1.0, 1.0)
    else:
        after = lower != notrans          # update cols beyond k
        transposed = not notrans
        order = range(nt) if after else range(nt - 1, -1, -1)
        rowpA = Panel(Panel.ROW, da, dt, dev)
        colpA = Panel(Panel.COL, da, dt, dev) if transposed else None
        colpB = Panel(Panel.COL, db, dt, dev)
        opB_upd = op if transposed else Op.NoTrans
        for k in order:
            kc = da.rank_of_tile_col(k)
            dtile = pan.bcast_diag_to_col(A, g, k, diag_ws)
            if db.rank_col == kc:
                inv = ops.tri_inverse_full(dtile, lower, unit)
                _right_mul_col_inplace(B, k, inv, op, 0, lrB)
                for li in range(lrB):
                    i = db.global_tile_of_local((li, 0))[0]
                    colpB.slot(li).copy_(B.tile((i, k)))
            _bcast_B_col(B, g, k, colpB, fill=False)
            lj0, lj1 = _prep_right_A_panel(A, g, k, after, transposed, rowpA, colpA)
            items = [(li, lj) for li in range(lrB) for lj in range(lj0, lj1)]
            _update_tiles(B, items, colpB, rowpA, Op.NoTrans, opB_upd, -1.0, 1.0)


# ---------------------------------------------------------------------------
# TRMM
# ---------------------------------------------------------------------------

def triangular_multiplication(side: Side, uplo: UpLo, op: Op, diag: Diag, alpha,
                              A: Matrix, B: Matrix, grid: Optional[CommGrid] = None) -> None:
    """In-place multiply B <- alpha op(A) B (Left) or alpha B op(A) (Right).

    Reference: ``multiplication/triangular/impl.h:203-726``. Same panel
    skeleton as TRSM with the loop direction reversed: each step broadcasts
    the pre-multiply B panel, accumulates the trailing contributions, then
    applies the diagonal-block triangular multiply.
    """
    _check_args(side, A, B)
    g = _trivial_grid(grid if grid is not None else B.grid)
    unit = diag == Diag.Unit
    lower = uplo == UpLo.Lower
    notrans = op == Op.NoTrans

    da, db = A.dist, B.dist
    nt = da.nr_tiles[0]
    nb = da.nb
    lrB, lcB = db.local_nr_tiles
    dev, dt = B.device, B.dtype
    diag_ws = torch.empty((nb, nb), dtype=dt, device=dev)

    if side == Side.Left:
        below = lower == notrans
        transposed = not notrans
        # each row's own step must precede the updates targeting it
        order = range(nt - 1, -1, -1) if below else range(nt)
        colpA = Panel(Panel.COL, da, dt, dev)
        rowpA = Panel(Panel.ROW, da, dt, dev) if transposed else None
        rowpB = Panel(Panel.ROW, db, dt, dev)
        opA_upd = op if transposed else Op.NoTrans
        for k in order:
            kr = da.rank_of_tile_row(k)
            dtile = pan.bcast_diag_to_row(A, g, k, diag_ws)
            _bcast_B_row(B, g, k, rowpB, fill=True)
            li0, li1 = _prep_left_A_panel(A, g, k, below, transposed, colpA, rowpA)
            items = [(li, lj) for li in range(li0, li1) for lj in range(lcB)]
            _update_tiles(B, items, colpA, rowpB, opA_upd, Op.NoTrans, 1.0, 1.0)
            if db.rank_row == kr:
                mask = ops.tri_mask(dtile, lower, unit)
                _row_from_panel_mul(B, k, rowpB, mask, op, 0, lcB)
    else:
        after = lower != notrans
        transposed = not notrans
        order = range(nt - 1, -1, -1) if after else range(nt)
        rowpA = Panel(Panel.ROW, da, dt, dev)
        colpA = Panel(Panel.COL, da, dt, dev) if transposed else None
        colpB = Panel(Panel.COL, db, dt, dev)
        opB_upd = op if transposed else Op.NoTrans
        for k in order:
            kc = da.rank_of_tile_col(k)
            dtile = pan.bcast_diag_to_col(A, g, k, diag_ws)
            _bcast_B_col(B, g, k, colpB, fill=True)
            lj0, lj1 = _prep_right_A_panel(A, g, k, after, transposed, rowpA, colpA)
            items = [(li, lj) for li in range(lrB) for lj in range(lj0, lj1)]
            _update_tiles(B, items, colpB, rowpA, Op.NoTrans, opB_upd, 1.0, 1.0)
            if db.rank_col == kc:
                mask = ops.tri_mask(dtile, lower, unit)
                _col_from_panel_mul(B, k, colpB, mask, op, 0, lrB)
    if alpha != 1:
        B.storage.mul_(alpha)
