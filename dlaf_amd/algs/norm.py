"""Max-norm of a distributed matrix.

Counterpart of ``auxiliary/norm/mc.h:1-163`` (per-tile lange/lantr max-abs +
MAX reduce to rank {0,0}); here the result is all-reduced so every rank
returns the same scalar (the reference's callers broadcast it anyway).
``uplo=Lower`` restricts to the lower triangle + diagonal (Hermitian
matrices); padding is excluded by using logical tile views.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..types import UpLo
from ..matrix.matrix import Matrix
from ..comm.grid import CommGrid
from ..comm import collectives as coll


def max_norm(mat: Matrix, uplo: UpLo = None, grid: Optional[CommGrid] = None) -> float:
    """Largest absolute element of the (triangle of the) distributed matrix."""
    d = mat.dist
    g = grid if grid is not None else mat.grid
    best = torch.zeros((), dtype=torch.float64, device=mat.device)
    for li, lj in d.iter_local_tiles():
        gi, gj = d.global_tile_of_local((li, lj))
        t = mat.tile_logical((gi, gj))
        if uplo == UpLo.Lower:
            if gi < gj:
                continue
            if gi == gj:
                t = torch.tril(t)
        elif uplo == UpLo.Upper:
            if gi > gj:
                continue
            if gi == gj:
                t = torch.triu(t)
        if t.numel():
            m = t.abs().max().to(torch.float64)
            best = torch.maximum(best, m)
    if g is not None and g.distributed:
        coll.all_reduce_max(best, g.full_group)
    return float(best.item())
