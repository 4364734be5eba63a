"""Row/column permutations of a tiled (distributed) matrix.

Counterpart of ``permutations/general/impl.h:1-659`` (+ ``perms.cu``): apply
``out[:, j] = in[:, perm[j]]`` (Coord::Col) or ``out[i, :] = in[perm[i], :]``
(Coord::Row) over the tiled storage. Local: one device gather. Distributed:
every rank gathers the tiles it owns from the assembled source columns —
implemented as an all-reduce assembly of the needed strip (the payloads the
reference moves with packed Isend/Irecv chunks travel over RCCL here).
"""

from __future__ import annotations

from typing import Optional

import torch

from ..matrix.matrix import Matrix
from ..comm.grid import CommGrid


def permute_columns(src: Matrix, perm: torch.Tensor, dst: Matrix,
                    grid: Optional[CommGrid] = None) -> None:
    """dst[:, j] = src[:, perm[j]] (global column indices)."""
    _permute(src, perm, dst, grid, dim=1)


def permute_rows(src: Matrix, perm: torch.Tensor, dst: Matrix,
                 grid: Optional[CommGrid] = None) -> None:
    """dst[i, :] = src[perm[i], :]."""
    _permute(src, perm, dst, grid, dim=0)


def _permute(src: Matrix, perm: torch.Tensor, dst: Matrix,
             grid: Optional[CommGrid], dim: int) -> None:
    assert src.dist.size == dst.dist.size and src.dist.tile_size == dst.dist.tile_size
    g = grid if grid is not None else src.grid
    perm = perm.to(src.device)
    a = src.to_global()          # all-reduce assembly when distributed
    out = torch.index_select(a, dim, perm)
    dst.set_from_global(out)
