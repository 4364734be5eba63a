"""Row/column permutations of a tiled (distributed) matrix.

Counterpart of ``permutations/general/impl.h:1-659`` (+ ``perms.cu``): apply
``out[:, j] = in[:, perm[j]]`` (Coord::Col) or ``out[i, :] = in[perm[i], :]``
(Coord::Row) over the tiled storage. Local: one device gather. Distributed
(round 2): PACKED PAIRWISE p2p along the permuted axis only — rank pair
(me -> q) exchanges one buffer holding my owned source lines that land in
q's owned destination lines (the reference's packed Isend/Irecv chunks,
``permutations/general/impl.h:303-321``); the orthogonal axis never moves.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

from ..matrix.matrix import Matrix
from ..comm.grid import CommGrid
from . import _pack


def permute_columns(src: Matrix, perm: torch.Tensor, dst: Matrix,
                    grid: Optional[CommGrid] = None) -> None:
    """dst[:, j] = src[:, perm[j]] (global column indices)."""
    _permute(src, perm, dst, grid, dim=1)


def permute_rows(src: Matrix, perm: torch.Tensor, dst: Matrix,
                 grid: Optional[CommGrid] = None) -> None:
    """dst[i, :] = src[perm[i], :]."""
    _permute(src, perm, dst, grid, dim=0)


def _cv(t: torch.Tensor) -> torch.Tensor:
    return torch.view_as_real(t) if t.is_complex() else t


def _permute(src: Matrix, perm: torch.Tensor, dst: Matrix,
             grid: Optional[CommGrid], dim: int) -> None:
    assert src.dist.size == dst.dist.size and \
        src.dist.tile_size == dst.dist.tile_size
    g = grid if grid is not None else src.grid
    m, n = src.dist.size
    permn = perm.detach().cpu().numpy().astype(np.int64)

    if g is None or not g.distributed:
        rows = np.arange(m)
        cols = np.arange(n)
        if dim == 1:
            blk = _pack.gather_block(src, rows, permn)
        else:
            blk = _pack.gather_block(src, permn, cols)
        _pack.scatter_block(dst, rows, cols, blk)
        return

    world = g.world_size
    me = g.rank
    my_pr, my_pc = me // g.grid_cols, me % g.grid_cols
    # the orthogonal axis stays put: exchange only with ranks sharing it
    if dim == 1:
        ortho_mine = _pack.owned_globals(src.dist, 0, m)
    else:
        ortho_mine = _pack.owned_globals(src.dist, 1, n)
    limit = n if dim == 1 else m
    src_lines = _pack.owned_globals(src.dist, dim, limit)
    dst_lines_mine = _pack.owned_globals(dst.dist, dim, limit)
    src_set = set(src_lines.tolist())

    for delta in range(world):
        to = (me + delta) % world
        fr = (me - delta) % world
        tpr, tpc = to // g.grid_cols, to % g.grid_cols
        fpr, fpc = fr // g.grid_cols, fr % g.grid_cols
        # send only to ranks sharing my orthogonal-axis ownership
        share_to = (tpr == my_pr) if dim == 1 else (tpc == my_pc)
        share_fr = (fpr == my_pr) if dim == 1 else (fpc == my_pc)
        # destination lines of `to`; their sources; which sources I own
        s_dst = (_pack.owned_globals_of_rank(dst.dist, dim, limit,
                                             tpc if dim == 1 else tpr)
                 if share_to else np.zeros(0, dtype=np.int64))
        s_dst = s_dst[np.isin(permn[s_dst], src_lines)] if s_dst.size else s_dst
        s_src = permn[s_dst] if s_dst.size else s_dst
        # lines I will receive from `fr`
        r_dst = dst_lines_mine if share_fr else np.zeros(0, dtype=np.int64)
        if r_dst.size:
            fr_src = _pack.owned_globals_of_rank(src.dist, dim, limit,
                                                 fpc if dim == 1 else fpr)
            r_dst = r_dst[np.isin(permn[r_dst], fr_src)]
        if delta == 0:
            if s_dst.size:
                if dim == 1:
                    blk = _pack.gather_block(src, ortho_mine, s_src)
                    _pack.scatter_block(dst, ortho_mine, s_dst, blk)
                else:
                    blk = _pack.gather_block(src, s_src, ortho_mine)
                    _pack.scatter_block(dst, s_dst, ortho_mine, blk)
            continue
        if dim == 1:
            sbuf = (_pack.gather_block(src, ortho_mine, s_src).contiguous()
                    if s_src.size else
                    torch.zeros(0, dtype=src.dtype, device=src.device))
            rshape = (ortho_mine.size, r_dst.size)
        else:
            sbuf = (_pack.gather_block(src, s_src, ortho_mine).contiguous()
                    if s_src.size else
                    torch.zeros(0, dtype=src.dtype, device=src.device))
            rshape = (r_dst.size, ortho_mine.size)
        rbuf = torch.zeros(rshape, dtype=dst.dtype, device=dst.device)
        reqs = []
        if rbuf.numel():
            reqs.append(dist.irecv(_cv(rbuf), src=fr, group=g.full_group))
        if sbuf.numel():
            reqs.append(dist.isend(_cv(sbuf), dst=to, group=g.full_group))
        for rq in reqs:
            rq.wait()
        if rbuf.numel():
            if dim == 1:
                _pack.scatter_block(dst, ortho_mine, r_dst, rbuf)
            else:
                _pack.scatter_block(dst, r_dst, ortho_mine, rbuf)
