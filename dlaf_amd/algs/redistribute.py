"""Matrix redistribution between tile sizes / source ranks.

Counterpart of the reference's redistribution miniapp
(``miniapp/miniapp_redistribution.cpp``) and retiling copy machinery: copy a
distributed matrix into a matrix with a different tile size on the same
process grid. Round 2: PACKED PAIRWISE p2p — each element travels exactly
once, rank pair (me -> q) exchanges one contiguous buffer holding the
intersection of my source ownership with q's destination ownership
(reference's packed-chunk Isend/Irecv, ``permutations/general/impl.h:
303-321``); no dense assembly, O(elements/p) memory per rank.
"""
from __future__ import annotations

import numpy as np
import torch
import torch.distributed as dist

from ..matrix.matrix import Matrix
from . import _pack


def _cv(t: torch.Tensor) -> torch.Tensor:
    return torch.view_as_real(t) if t.is_complex() else t


def redistribute(src: Matrix, dst: Matrix) -> None:
    assert src.dist.size == dst.dist.size, (src.dist.size, dst.dist.size)
    g = src.grid if src.grid is not None else dst.grid
    m, n = src.dist.size
    if g is None or not g.distributed:
        my_r = np.arange(m)
        my_c = np.arange(n)
        _pack.scatter_block(dst, my_r, my_c,
                            _pack.gather_block(src, my_r, my_c))
        return
    assert (src.dist.grid_rows, src.dist.grid_cols) == \
        (dst.dist.grid_rows, dst.dist.grid_cols)
    world = g.world_size
    me = g.rank

    src_r = _pack.owned_globals(src.dist, 0, m)
    src_c = _pack.owned_globals(src.dist, 1, n)

    def coords(r):
        return r // g.grid_cols, r % g.grid_cols

    my_pr, my_pc = coords(me)
    dst_r_mine = _pack.owned_globals(dst.dist, 0, m)
    dst_c_mine = _pack.owned_globals(dst.dist, 1, n)

    for delta in range(world):
        to = (me + delta) % world
        fr = (me - delta) % world
        tpr, tpc = coords(to)
        s_rows = np.intersect1d(src_r,
                                _pack.owned_globals_of_rank(dst.dist, 0, m, tpr))
        s_cols = np.intersect1d(src_c,
                                _pack.owned_globals_of_rank(dst.dist, 1, n, tpc))
        fpr, fpc = coords(fr)
        r_rows = np.intersect1d(dst_r_mine,
                                _pack.owned_globals_of_rank(src.dist, 0, m, fpr))
        r_cols = np.intersect1d(dst_c_mine,
                                _pack.owned_globals_of_rank(src.dist, 1, n, fpc))
        if delta == 0:
            if s_rows.size and s_cols.size:
                _pack.scatter_block(dst, s_rows, s_cols,
                                    _pack.gather_block(src, s_rows, s_cols))
            continue
        sbuf = (_pack.gather_block(src, s_rows, s_cols).contiguous()
                if s_rows.size and s_cols.size
                else torch.zeros(0, dtype=src.dtype, device=src.device))
        rbuf = torch.zeros((r_rows.size, r_cols.size), dtype=dst.dtype,
                           device=dst.device)
        reqs = []
        if rbuf.numel():
            reqs.append(dist.irecv(_cv(rbuf), src=fr, group=g.full_group))
        if sbuf.numel():
            reqs.append(dist.isend(_cv(sbuf), dst=to, group=g.full_group))
        for rq in reqs:
            rq.wait()
        if rbuf.numel():
            _pack.scatter_block(dst, r_rows, r_cols, rbuf)
