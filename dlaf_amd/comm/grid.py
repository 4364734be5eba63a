"""2D process grid over torch.distributed (RCCL on GPU / gloo on CPU).

Counterpart of the reference's ``CommunicatorGrid`` (``communication/
communicator_grid.h:37-158``): builds row/col/full communicators from the world
communicator with row-major rank order, and exposes per-direction communication
"chains". On MI355X the chain is a dedicated HIP stream per direction: RCCL (like
MPI in the reference) requires collectives on one communicator to be issued in the
same order on every rank, and issuing them from a single per-direction stream in
deterministic program order provides exactly the ordering guarantee of the
reference's ``CommunicatorPipeline::exclusive()``.

A ``CommGrid`` can also be constructed without torch.distributed initialized, in
which case it is the trivial 1x1 grid (local-only algorithms).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist


class CommGrid:
    def __init__(self, grid_rows: int = 1, grid_cols: int = 1, device: Optional[torch.device] = None):
        self.grid_rows = grid_rows
        self.grid_cols = grid_cols
        if dist.is_available() and dist.is_initialized():
            self.world_size = dist.get_world_size()
            self.rank = dist.get_rank()
            assert grid_rows * grid_cols == self.world_size, (
                f"grid {grid_rows}x{grid_cols} != world size {self.world_size}"
            )
        else:
            self.world_size = 1
            self.rank = 0
            assert grid_rows == 1 and grid_cols == 1, "distributed grid needs torch.distributed"

        # row-major rank order (reference: communicator_pipeline.h:26)
        self.rank_row = self.rank // grid_cols
        self.rank_col = self.rank % grid_cols

        self._row_group = None
        self._col_group = None
        self._full_group = None
        if self.world_size > 1:
            # dist.new_group must be called identically on all ranks for every group.
            row_groups = []
            for r in range(grid_rows):
                ranks = [r * grid_cols + c for c in range(grid_cols)]
                row_groups.append(dist.new_group(ranks=ranks))
            col_groups = []
            for c in range(grid_cols):
                ranks = [r * grid_cols + c for r in range(grid_rows)]
                col_groups.append(dist.new_group(ranks=ranks))
            self._row_group = row_groups[self.rank_row]
            self._col_group = col_groups[self.rank_col]
            self._full_group = dist.group.WORLD

    # ---- rank math ----
    def rank_full(self, rank_row: int, rank_col: int) -> int:
        return rank_row * self.grid_cols + rank_col

    @property
    def my_rank(self) -> Tuple[int, int]:
        return (self.rank_row, self.rank_col)

    @property
    def distributed(self) -> bool:
        return self.world_size > 1

    # ---- groups (None means single-member: no comm needed) ----
    @property
    def row_group(self):
        """Communicator over the ranks of my grid ROW (varying col). Size grid_cols."""
        return self._row_group if self.grid_cols > 1 else None

    @property
    def col_group(self):
        """Communicator over the ranks of my grid COLUMN (varying row). Size grid_rows."""
        return self._col_group if self.grid_rows > 1 else None

    @property
    def full_group(self):
        return self._full_group

    def group_rank_in_row(self, rank_col: int) -> int:
        """Group-local rank of grid column ``rank_col`` inside the row group."""
        return rank_col

    def group_rank_in_col(self, rank_row: int) -> int:
        return rank_row

    def global_rank_of_row_member(self, rank_col: int) -> int:
        return self.rank_full(self.rank_row, rank_col)

    def global_rank_of_col_member(self, rank_row: int) -> int:
        return self.rank_full(rank_row, self.rank_col)

    def barrier(self):
        if self.distributed:
            dist.barrier()
