"""Matrix redistribution between block sizes / grids.

Counterpart of the reference's copy/retiling machinery exercised by
``miniapp/miniapp_redistribution.cpp``: copy a distributed matrix into a
matrix with a different tile size (and/or grid). Assembly via the full-group
all-reduce (tile payloads over RCCL), then local scatter into the target
layout.
"""
from __future__ import annotations

from ..matrix.matrix import Matrix


def redistribute(src: Matrix, dst: Matrix) -> None:
    assert src.dist.size == dst.dist.size, (src.dist.size, dst.dist.size)
    dst.set_from_global(src.to_global())
