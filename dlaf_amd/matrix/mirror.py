"""MatrixMirror, MatrixRef and matrix dump/restore.

Counterparts of the reference's ``matrix/matrix_mirror.h`` (CPU<->GPU bridge),
``matrix/matrix_ref.h`` (tile-aligned sub-matrix views used for
partial-spectrum back-transforms) and ``matrix/hdf5.h`` (debug dump/restore;
HDF5 is not in this image, so the dump format is a torch checkpoint with the
same role — reproducing solver inputs/outputs for debugging).
"""

from __future__ import annotations

from typing import Tuple

import torch

from ..core.distribution import Distribution
from .matrix import Matrix


class MatrixMirror:
    """Copy a matrix to a target device on entry, back on exit (no-op when
    the devices match). Usage::

        with MatrixMirror(host_mat, "cuda") as dev_mat:
            cholesky_factorization(UpLo.Lower, dev_mat)
    """

    def __init__(self, source: Matrix, device):
        self.source = source
        self.device = torch.device(device)
        if self.device == source.device:
            self.target = source
        else:
            self.target = Matrix(source.dist, source.dtype, self.device, source.grid)
            self._stage_copy(source.storage, self.target.storage)

    @staticmethod
    def _stage_copy(src: torch.Tensor, dst: torch.Tensor) -> None:
        """H2D/D2H through a pooled PINNED bounce buffer (async-capable DMA;
        an unpinned host tensor forces a slow pageable copy). Device->device
        and host->host copies go direct."""
        if src.device.type == dst.device.type:
            dst.copy_(src)
            return
        from ..runtime import memory as mempool
        host = src if src.device.type == "cpu" else dst
        if host.is_pinned():
            dst.copy_(src)
            return
        buf = mempool.acquire(src.shape, src.dtype, torch.device("cpu"),
                              pinned=True)
        try:
            if src.device.type == "cpu":
                buf.copy_(src)
                dst.copy_(buf, non_blocking=True)
                torch.cuda.synchronize(dst.device)
            else:
                buf.copy_(src)
                dst.copy_(buf)
        finally:
            mempool.release(buf)

    def get(self) -> Matrix:
        return self.target

    def __enter__(self) -> Matrix:
        return self.target

    def __exit__(self, *exc):
        self.copy_back()
        return False

    def copy_back(self) -> None:
        if self.target is not self.source:
            self._stage_copy(self.target.storage, self.source.storage)


class MatrixRef:
    """Tile-aligned sub-matrix view over a parent Matrix.

    The reference allows element-aligned origins (``matrix_ref.h:36-281``);
    the tiled-storage design here requires tile alignment, which covers every
    in-tree use (partial-spectrum column slices, D&C sub-problem GEMMs).
    """

    def __init__(self, parent: Matrix, origin: Tuple[int, int], size: Tuple[int, int]):
        d = parent.dist
        assert origin[0] % d.mb == 0 and origin[1] % d.nb == 0, "tile-aligned origins only"
        assert origin[0] + size[0] <= d.m and origin[1] + size[1] <= d.n
        self.parent = parent
        self.origin = origin
        self.size = size
        self.tile_origin = (origin[0] // d.mb, origin[1] // d.nb)

    @property
    def dist(self) -> Distribution:
        d = self.parent.dist
        return Distribution(self.size[0], self.size[1], d.mb, d.nb,
                            d.grid_rows, d.grid_cols, d.rank_row, d.rank_col,
                            d.rank_of_tile_row(self.tile_origin[0]),
                            d.rank_of_tile_col(self.tile_origin[1]))

    def tile(self, gtile: Tuple[int, int]) -> torch.Tensor:
        return self.parent.tile((gtile[0] + self.tile_origin[0],
                                 gtile[1] + self.tile_origin[1]))

    def to_global(self) -> torch.Tensor:
        full = self.parent.to_global()
        r0, c0 = self.origin
        return full[r0:r0 + self.size[0], c0:c0 + self.size[1]]


def save_matrix(mat: Matrix, path: str, name: str = "matrix") -> None:
    """Dump the global matrix (debug/repro mechanism, reference hdf5.h role)."""
    torch.save({name: mat.to_global().cpu(),
                "tile": mat.dist.tile_size, "size": mat.dist.size}, path)


def load_matrix(path: str, mat: Matrix, name: str = "matrix") -> None:
    data = torch.load(path, weights_only=True)
    mat.set_from_global(data[name].to(mat.device))
