"""Distributed tiled matrix.

Counterpart of the reference's ``Matrix<T, D>`` + tiles-allocation layout
(``include/dlaf/matrix/matrix.h``, ``matrix/allocation.h`` AllocationLayout::Tiles):
the local part of a 2D block-cyclic matrix is ONE contiguous torch tensor of shape
``[local_tile_rows, local_tile_cols, mb, nb]`` so that

* every tile is a contiguous ``mb x nb`` row-major block — the unit of RCCL
  messages and of the fused HIP kernels' descriptor math;
* a run of tiles along a local tile-row is a constant-stride sequence (stride
  ``mb*nb`` tiles along the column index), which the fused kernels exploit for
  K-loops.

Padding invariant ("identity extension"): partial edge tiles are stored padded to
the full ``mb x nb``; the padded region always holds the identity extension of the
matrix (zeros, and ones on the global diagonal for square matrices). GEMM / SYRK /
TRSM / POTRF / TRTRI / HEGST on full padded tiles are then *exact* on the logical
matrix — no edge masking is needed anywhere in the BLAS-3 path. Generators
establish the invariant; extraction (``to_global``) slices it away.

Unlike the reference there is no per-tile future/``async_rw_mutex`` object:
dependency tracking is done with HIP streams and events at algorithm phase
granularity (see ``runtime/streams.py``).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..core.distribution import Distribution
from ..comm.grid import CommGrid
from ..comm import collectives as coll

Coord = Tuple[int, int]


class Matrix:
    def __init__(
        self,
        dist_: Distribution,
        dtype: torch.dtype = torch.float64,
        device: Optional[torch.device] = None,
        grid: Optional[CommGrid] = None,
        _storage: Optional[torch.Tensor] = None,
    ):
        if grid is not None:
            assert (grid.grid_rows, grid.grid_cols) == (dist_.grid_rows, dist_.grid_cols)
            if dist_.rank_row != grid.rank_row or dist_.rank_col != grid.rank_col:
                dist_ = dist_.for_rank(grid.rank_row, grid.rank_col)
        self.dist = dist_
        self.grid = grid
        self.dtype = dtype
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        lr, lc = dist_.local_nr_tiles
        if _storage is not None:
            assert _storage.shape == (lr, lc, dist_.mb, dist_.nb)
            self.storage = _storage
        else:
            self.storage = torch.zeros(
                (max(lr, 1), max(lc, 1), dist_.mb, dist_.nb), dtype=dtype, device=self.device
            )[:lr, :lc]

    # ---- constructors ----
    @classmethod
    def create(
        cls,
        m: int,
        n: int,
        mb: int,
        nb: int,
        dtype: torch.dtype = torch.float64,
        device=None,
        grid: Optional[CommGrid] = None,
    ) -> "Matrix":
        g = grid
        if g is None:
            d = Distribution(m, n, mb, nb)
        else:
            d = Distribution(m, n, mb, nb, g.grid_rows, g.grid_cols, g.rank_row, g.rank_col)
        return cls(d, dtype=dtype, device=device, grid=g)

    def like(self, m=None, n=None, mb=None, nb=None, dtype=None) -> "Matrix":
        """A new zero matrix with the same grid/device, optionally reshaped."""
        d = self.dist
        return Matrix.create(
            d.m if m is None else m,
            d.n if n is None else n,
            d.mb if mb is None else mb,
            d.nb if nb is None else nb,
            dtype=self.dtype if dtype is None else dtype,
            device=self.device,
            grid=self.grid,
        )

    # ---- tile access (local rank only) ----
    def tile(self, gtile: Coord) -> torch.Tensor:
        """Full padded mb x nb view of a locally-owned global tile."""
        li, lj = self.dist.local_tile_of_global(gtile)
        return self.storage[li, lj]

    def tile_offset(self, gtile: Coord) -> int:
        """Element offset of a locally-owned tile inside the flat local storage
        (feeds the fused-kernel GemmDescs)."""
        li, lj = self.dist.local_tile_of_global(gtile)
        return self.local_tile_offset(li, lj)

    def local_tile_offset(self, li: int, lj: int) -> int:
        lc = self.dist.local_nr_tiles[1]
        return (li * lc + lj) * self.dist.mb * self.dist.nb

    def local_tile(self, ltile: Coord) -> torch.Tensor:
        return self.storage[ltile[0], ltile[1]]

    def tile_logical(self, gtile: Coord) -> torch.Tensor:
        """Logical (unpadded) view of a locally-owned global tile."""
        ts = self.dist.tile_size_of(gtile)
        return self.tile(gtile)[: ts[0], : ts[1]]

    # ---- whole-matrix helpers (tests / small problems) ----
    def set_zero(self) -> None:
        self.storage.zero_()

    def set_from_global(self, a: torch.Tensor) -> None:
        """Fill the local part from a replicated global [m, n] tensor."""
        assert a.shape == (self.dist.m, self.dist.n)
        a = a.to(device=self.device, dtype=self.dtype)
        self.storage.zero_()
        d = self.dist
        for li, lj in d.iter_local_tiles():
            gi, gj = d.global_tile_of_local((li, lj))
            r0, c0 = gi * d.mb, gj * d.nb
            ts = d.tile_size_of((gi, gj))
            self.storage[li, lj, : ts[0], : ts[1]] = a[r0 : r0 + ts[0], c0 : c0 + ts[1]]
        self._set_identity_pad()

    def _set_identity_pad(self) -> None:
        """Establish the identity-extension invariant on partial edge tiles."""
        d = self.dist
        if d.m % d.mb == 0 and d.n % d.nb == 0:
            return
        for li, lj in d.iter_local_tiles():
            gi, gj = d.global_tile_of_local((li, lj))
            ts = d.tile_size_of((gi, gj))
            t = self.storage[li, lj]
            if ts[0] < d.mb:
                t[ts[0] :, :] = 0
            if ts[1] < d.nb:
                t[:, ts[1] :] = 0
            # ones on the global diagonal inside the padded region (square tiles
            # on the diagonal of a square matrix)
            if gi * d.mb == gj * d.nb and d.mb == d.nb:
                for p in range(max(ts[0], ts[1]), d.mb):
                    if p >= ts[0] and p >= ts[1]:
                        t[p, p] = 1

    def to_global(self) -> torch.Tensor:
        """Assemble the full logical [m, n] matrix on every rank (tests only)."""
        d = self.dist
        out = torch.zeros((d.m, d.n), dtype=self.dtype, device=self.device)
        for li, lj in d.iter_local_tiles():
            gi, gj = d.global_tile_of_local((li, lj))
            r0, c0 = gi * d.mb, gj * d.nb
            ts = d.tile_size_of((gi, gj))
            out[r0 : r0 + ts[0], c0 : c0 + ts[1]] = self.storage[li, lj, : ts[0], : ts[1]]
        if self.grid is not None and self.grid.distributed:
            coll.all_reduce_sum(out, self.grid.full_group)
        return out

    def copy_(self, other: "Matrix") -> None:
        assert self.dist.size == other.dist.size and self.dist.tile_size == other.dist.tile_size
        self.storage.copy_(other.storage)

    def clone(self) -> "Matrix":
        return Matrix(self.dist, self.dtype, self.device, self.grid, _storage=self.storage.clone())

    def __repr__(self) -> str:
        d = self.dist
        return (
            f"Matrix({d.m}x{d.n}, tile {d.mb}x{d.nb}, grid {d.grid_rows}x{d.grid_cols}, "
            f"rank ({d.rank_row},{d.rank_col}), {self.dtype}, {self.device})"
        )
