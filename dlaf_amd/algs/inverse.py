"""Triangular inverse (TRTRI) and inverse from Cholesky factor (POTRI).

Counterpart of ``inverse/triangular/impl.h:183-549`` and
``inverse/cholesky/impl.h:180-540``.

TRTRI (Lower, in place, X = L^-1), processed by tile-column k DESCENDING
(LAPACK xTRTRI lower order): the trailing inverse X22 (already computed, in
place) is applied to the original column L[>k, k]

    acc[i]  = sum_{k < j <= i} X[i,j] L[j,k]      (fused GEMMs + row-reduce)
    X[i,k]  = -acc[i] @ inv(L[k,k])
    X[k,k]  = inv(L[k,k])

LAUUM-stage of POTRI (A^-1 = X^H X for X = L^-1, lower triangle, in place),
processed by tile-row k ASCENDING: row k of X is broadcast (direct row panel +
two-hop transposed panel), then contributes the rank-nb update
B[i,j] += X[k,i]^H X[k,j] for j <= i <= k; the first contribution to row i
arrives exactly at step k = i and overwrites X there (beta = 0).

Comm per step: contiguous panel broadcasts + one row-direction reduce
(TRTRI); all local tile products are ONE fused-GEMM launch on GPU.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..types import UpLo, Diag, Op, is_complex
from ..matrix.matrix import Matrix
from ..matrix.panel import Panel
from ..comm.grid import CommGrid
from ..comm import collectives as coll
from ..ops import tile_ops as ops
from . import _panels as pan
from .triangular import _trivial_grid


def _opc(dtype) -> Op:
    return Op.ConjTrans if is_complex(dtype) else Op.Trans


def triangular_inverse(uplo: UpLo, diag: Diag, mat: Matrix,
                       grid: Optional[CommGrid] = None) -> None:
    """In-place inverse of a (lower) triangular tiled matrix.

    Reference: ``inverse/triangular/impl.h`` (Lower; Upper by symmetry is not
    provided by the reference miniapps either).
    """
    if uplo == UpLo.Upper:
        from ._uplo import transpose_storage
        transpose_storage(mat)
        triangular_inverse(UpLo.Lower, diag, mat, grid)
        transpose_storage(mat)
        return
    d = mat.dist
    assert d.m == d.n and d.mb == d.nb
    g = _trivial_grid(grid if grid is not None else mat.grid)
    unit = diag == Diag.Unit
    if (mat.device.type == "cuda" and not g.distributed
            and d.nr_tiles[0] > 2):
        _trtri_local_gpu(mat, unit)
        return
    nt = d.nr_tiles[0]
    nb = d.nb
    dev, dt = mat.device, mat.dtype
    lr, lc = d.local_nr_tiles

    colpL = Panel(Panel.COL, d, dt, dev)    # original column k, keyed by rows
    rowpL = Panel(Panel.ROW, d, dt, dev)    # same tiles keyed by cols (transposed)
    acc = Panel(Panel.COL, d, dt, dev)      # partial sums of X22 @ L[>k,k]
    diag_ws = torch.empty((nb, nb), dtype=dt, device=dev)

    # Unit diag: the diagonal is implicit (ones) and must not be modified; the
    # algorithm stores explicit unit-inverses (ones on the diagonal) so the
    # trailing products are plain GEMMs, and the caller's diagonal entries are
    # restored afterwards.
    saved_diag = {}
    if unit:
        for k in range(nt):
            if d.is_tile_local((k, k)):
                saved_diag[k] = mat.tile((k, k)).diagonal().clone()

    # hoist ALL diagonal-block inversions out of the sequential k-loop: they
    # only read original L diagonal tiles (never touched before step k), and
    # on GPU each tri_inverse_full is a multi-launch recursion that otherwise
    # sits on the critical path of every step
    invs = {}
    own, own_tiles = [], []
    for k in range(nt - 1, -1, -1):
        dtile = pan.bcast_diag_to_col(mat, g, k, diag_ws)
        if d.rank_col == d.rank_of_tile_col(k):
            own.append(k)
            own_tiles.append(dtile.clone())
    for k, inv in zip(own, ops.tri_inverse_full_many(own_tiles, lower=True,
                                                     unit=unit)):
        invs[k] = inv

    for k in range(nt - 1, -1, -1):
        kc = d.rank_of_tile_col(k)
        li0 = d.next_local_tile_row(k + 1)
        lj0 = d.next_local_tile_col(k + 1)
        # 1) acc[i] = sum_{k<j<=i} X[i,j] L[j,k]  (X = already-inverted trailing).
        # NOTE: the panel broadcasts must run on every rank of their process
        # group — their internal ranges are group-consistent, the local row
        # range li0:lr is NOT (it differs across the rows of a column group).
        pan.bcast_col_panel(mat, g, k, li0, lr, colpL)
        pan.transpose_col_to_row(d, g, colpL, rowpL, lj0, lc)
        if li0 < lr:
            acc.storage[li0:lr].zero_()
            if mat.device.type == "cuda":
                # one desc per output row with ktiles accumulation: the k-sum
                # over j MUST live inside a single desc — descs sharing a C
                # block race (each workgroup does a non-atomic C += update)
                rows6 = []
                ts = nb * nb
                for li in range(li0, lr):
                    i = d.global_tile_of_local((li, 0))[0]
                    cnt = d.next_local_tile_col(i + 1) - lj0
                    if cnt > 0:
                        rows6.append([acc.offset(li),
                                      mat.local_tile_offset(li, lj0),
                                      rowpL.offset(lj0), cnt, ts, ts])
                if rows6:
                    ops.gemm_fused(acc.storage, mat.storage, rowpL.storage,
                                   np.array(rows6, dtype=np.int64),
                                   nb, nb, nb, nb, nb, nb,
                                   Op.NoTrans, Op.NoTrans, 1.0, 1.0)
            else:
                items = []
                for li in range(li0, lr):
                    i = d.global_tile_of_local((li, 0))[0]
                    for lj in range(lj0, d.next_local_tile_col(i + 1)):
                        items.append((acc.offset(li), mat.local_tile_offset(li, lj),
                                      rowpL.offset(lj)))
                ops.gemm_items(acc.storage, mat.storage, rowpL.storage, items, nb,
                               Op.NoTrans, Op.NoTrans, 1.0, 1.0)
            if g.row_group is not None:
                coll.reduce_sum(acc.range_view(li0, lr), g.global_rank_of_row_member(kc),
                                g.row_group)
        # 2) apply the precomputed diagonal-block inverse, write column k
        if d.rank_col == kc:
            inv = invs[k]
            if li0 < lr:
                # X[i,k] = -acc[i] @ inv  (write into matrix tiles)
                c = [mat.tile_offset((d.global_tile_of_local((li, 0))[0], k))
                     for li in range(li0, lr)]
                items = list(zip(c, [acc.offset(li) for li in range(li0, lr)], [0] * len(c)))
                ops.gemm_items(mat.storage, acc.storage, inv, items, nb,
                               Op.NoTrans, Op.NoTrans, -1.0, 0.0, uniform=True)
            kr = d.rank_of_tile_row(k)
            if d.rank_row == kr:
                mat.tile((k, k)).copy_(inv)

    if unit:
        for k, dg in saved_diag.items():
            mat.tile((k, k)).diagonal().copy_(dg)


def _trtri_local_gpu(mat: Matrix, unit: bool) -> None:
    """Single-GPU TRTRI with the POTRF/TRSM-style two-stream lookahead.

    Descending column loop X[:,k] = -inv(L_kk) ... with the k-sum split so
    the O(nt^2) "partial" accumulation of step k (tiles j >= k+2, available
    two steps early) runs on the update stream while the critical chain
    (the j = k+1 fix-up + the column write) stays on the high-priority
    stream — the round-1 backlog's sequential acc-chain is now one small
    GEMM per step. Round-1: 23.9 TF at N=16384; reference counterpart
    ``inverse/triangular/impl.h:183-549``.
    """
    from ..runtime.streams import get_runtime

    d = mat.dist
    nt = d.nr_tiles[0]
    nb = d.nb
    dev, dt = mat.device, mat.dtype
    ts = nb * nb
    st = mat.storage  # [nt, nt, nb, nb]

    saved_diag = {}
    if unit:
        for k in range(nt):
            saved_diag[k] = mat.tile((k, k)).diagonal().clone()

    # hoisted diagonal-block inversions (never touched before their step);
    # full padded tiles (identity pad) so every desc is nb-uniform
    own_tiles = [st[k, k].clone() for k in range(nt)]
    invs = ops.tri_inverse_full_many(own_tiles, lower=True, unit=unit)
    inv_stack = torch.stack(invs)  # [nt, nb, nb]

    # K-split: a row's k-sum is chunked over P accumulator planes so the
    # fused launch is load-balanced (one desc per (row, chunk) instead of
    # one triangular-length chain per row — the chain made every launch as
    # long as its LONGEST row and capped round-1 TRTRI at ~23 TF)
    import os
    kchunk = int(os.environ.get("DLAF_TRTRI_KSPLIT", "4"))
    P = max(1, -(-(nt - 2) // kchunk))
    acc = torch.zeros((2, P, nt, nb, nb), dtype=dt, device=dev)
    accsum = torch.zeros((nt, nb, nb), dtype=dt, device=dev)

    rt = get_runtime(dev)
    sp, su = rt.hp_streams[0], rt.np_streams[0]
    cur = torch.cuda.current_stream(dev)
    sp.wait_stream(cur)
    su.wait_stream(cur)
    ev_write = [None] * (nt + 1)
    ev_partial = [None] * (nt + 1)

    def tile_off(i, j):
        return (i * nt + j) * ts

    # su: partial_k = sum_{j >= k+2} X[:, j] L[j, k], K-split into planes:
    # desc per (row i, chunk c) -> acc[buf, c, i] (distinct C blocks)
    def launch_partial(k, buf):
        rows = []
        for i in range(k + 2, nt):
            cnt = i - (k + 1)
            for c in range(0, cnt, kchunk):
                cc = min(kchunk, cnt - c)
                rows.append([(buf * P + c // kchunk) * nt * ts + i * ts,
                             tile_off(i, k + 2 + c),
                             tile_off(k + 2 + c, k), cc, ts, nt * ts])
        acc[buf].zero_()  # short rows leave high planes untouched
        if rows:
            ops.gemm_fused(acc.view(-1), st.view(-1), st.view(-1),
                           np.array(rows, dtype=np.int64), nb, nb, nb,
                           nb, nb, nb, Op.NoTrans, Op.NoTrans, 1.0, 0.0)
        e = torch.cuda.Event()
        e.record(su)
        return e

    for k in range(nt - 1, -1, -1):
        buf = k & 1
        # partial for this step was launched two iterations ago; launch the
        # next partial (k-2) as soon as this step's column write ordering
        # allows reading columns >= k (ev_write[k] below on sp; partial k-2
        # reads cols >= k — but column k is written THIS step, so it must
        # wait for that write; chain via events)
        with torch.cuda.stream(sp):
            if ev_partial[k] is not None:
                sp.wait_event(ev_partial[k])
            n_rows = nt - (k + 1)
            if n_rows > 0:
                # collapse the K-split planes, then fix-up + column write
                torch.sum(acc[buf], dim=0, out=accsum)
                fix = [[i * ts, tile_off(i, k + 1),
                        tile_off(k + 1, k), 1, 0, 0]
                       for i in range(k + 1, nt)]
                ops.gemm_fused(accsum.view(-1), st.view(-1), st.view(-1),
                               np.array(fix, dtype=np.int64), nb, nb, nb,
                               nb, nb, nb, Op.NoTrans, Op.NoTrans, 1.0, 1.0)
                # column write: X[:, k] = -acc @ inv_k
                wr = [[tile_off(i, k), i * ts, k * ts, 1, 0, 0]
                      for i in range(k + 1, nt)]
                ops.gemm_fused(st.view(-1), accsum.view(-1), inv_stack.view(-1),
                               np.array(wr, dtype=np.int64), nb, nb, nb,
                               nb, nb, nb, Op.NoTrans, Op.NoTrans, -1.0, 0.0)
            t = mat.tile((k, k))
            t.copy_(inv_stack[k][:t.shape[0], :t.shape[1]])
            e = torch.cuda.Event()
            e.record(sp)
            ev_write[k] = e
        if k - 2 >= 0:
            with torch.cuda.stream(su):
                su.wait_event(ev_write[k])          # partial reads col k
                if k < nt - 1 and ev_write[k + 1] is not None:
                    su.wait_event(ev_write[k + 1])
                ev_partial[k - 2] = launch_partial(k - 2, (k - 2) & 1)

    cur.wait_stream(sp)
    cur.wait_stream(su)

    if unit:
        for k, dg in saved_diag.items():
            mat.tile((k, k)).diagonal().copy_(dg)


def inverse_from_cholesky_factor(uplo: UpLo, mat: Matrix,
                                 grid: Optional[CommGrid] = None) -> None:
    """A^-1 = L^-H L^-1 from the Cholesky factor L, lower triangle in place.

    Reference: ``inverse/cholesky/impl.h:180-540`` (TRTRI then LAUUM-style
    assembly).
    """
    if uplo == UpLo.Upper:
        from ._uplo import transpose_storage
        transpose_storage(mat)
        inverse_from_cholesky_factor(UpLo.Lower, mat, grid)
        transpose_storage(mat)
        return
    d = mat.dist
    assert d.m == d.n and d.mb == d.nb
    g = _trivial_grid(grid if grid is not None else mat.grid)
    triangular_inverse(UpLo.Lower, Diag.NonUnit, mat, grid)

    nt = d.nr_tiles[0]
    nb = d.nb
    dev, dt = mat.device, mat.dtype
    lr, lc = d.local_nr_tiles
    opc = _opc(dt)
    rowpX = Panel(Panel.ROW, d, dt, dev)   # X[k, j] keyed by local cols
    colpX = Panel(Panel.COL, d, dt, dev)   # X[k, i] keyed by local rows

    for k in range(nt):
        # broadcast row k of X (cols <= k), direct + transposed
        lj1 = d.next_local_tile_col(k + 1)
        pan.bcast_row_panel(mat, g, k, 0, lj1, rowpX)
        li1 = d.next_local_tile_row(k + 1)
        pan.transpose_row_to_col(d, g, rowpX, colpX, 0, li1)
        # B[i,j] (+)= X[k,i]^H X[k,j] for j <= i <= k; first write at i == k
        first, accum = [], []
        for li in range(0, li1):
            i = d.global_tile_of_local((li, 0))[0]
            for lj in range(0, d.next_local_tile_col(i + 1)):
                trip = (mat.local_tile_offset(li, lj), colpX.offset(li), rowpX.offset(lj))
                (first if i == k else accum).append(trip)
        ops.gemm_items(mat.storage, colpX.storage, rowpX.storage, first, nb,
                       opc, Op.NoTrans, 1.0, 0.0, uniform=True)
        ops.gemm_items(mat.storage, colpX.storage, rowpX.storage, accum, nb,
                       opc, Op.NoTrans, 1.0, 1.0, uniform=True)
