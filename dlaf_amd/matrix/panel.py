"""Panel workspace: a 1D strip of tiles aligned with a matrix distribution.

Counterpart of the reference's ``Panel<axis, T, D>`` (``matrix/panel.h:42-632``),
simplified: storage is one contiguous ``[nslots, mb, nb]`` tensor (slot = local
tile row for a column panel / local tile col for a row panel), so a contiguous
slot range is ONE RCCL broadcast and slot offsets feed the fused-GEMM
descriptors directly.
"""

from __future__ import annotations

import torch

from ..core.distribution import Distribution


class Panel:
    COL = "col"  # one tile per local tile-ROW (a column of tiles)
    ROW = "row"  # one tile per local tile-COL (a row of tiles)

    def __init__(self, axis: str, dist: Distribution, dtype, device):
        self.axis = axis
        self.dist = dist
        lr, lc = dist.local_nr_tiles
        self.nslots = lr if axis == Panel.COL else lc
        self.mb, self.nb = dist.mb, dist.nb
        self.storage = torch.zeros(
            (max(self.nslots, 1), self.mb, self.nb), dtype=dtype, device=device
        )

    def slot(self, i: int) -> torch.Tensor:
        return self.storage[i]

    def offset(self, i: int) -> int:
        return i * self.mb * self.nb

    def range_view(self, start: int, stop: int) -> torch.Tensor:
        """Contiguous view over slots [start, stop) — a single collective payload."""
        return self.storage[start:stop]
