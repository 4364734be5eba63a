"""Packed sub-matrix gather/scatter over tiled block-cyclic storage.

The building block of the p2p permutation / redistribution exchanges
(reference ``permutations/general/impl.h:303-321``): arbitrary global
(rows, cols) index lists are mapped to (local tile, offset) pairs and
gathered/scattered with ONE advanced-indexing op — no dense assembly.
"""

from __future__ import annotations

import numpy as np
import torch

from ..core import index as ix


def _coords(dist, gl: np.ndarray, axis: int):
    """(local tile idx, in-tile offset) tensors for global element indices."""
    if axis == 0:
        b, grid = dist.mb, dist.grid_rows
    else:
        b, grid = dist.nb, dist.grid_cols
    t = gl // b
    lt = t // grid  # local tile index of a tile this rank owns
    off = gl % b
    return torch.from_numpy(lt), torch.from_numpy(off)


def gather_block(mat, rows: np.ndarray, cols: np.ndarray) -> torch.Tensor:
    """[len(rows), len(cols)] block of globally-indexed elements, all of
    which must be owned by this rank."""
    d = mat.dist
    st = mat.storage
    ltr, offr = _coords(d, rows, 0)
    ltc, offc = _coords(d, cols, 1)
    dev = st.device
    return st[ltr.to(dev)[:, None], ltc.to(dev)[None, :],
              offr.to(dev)[:, None], offc.to(dev)[None, :]]


def scatter_block(mat, rows: np.ndarray, cols: np.ndarray,
                  block: torch.Tensor) -> None:
    d = mat.dist
    st = mat.storage
    ltr, offr = _coords(d, rows, 0)
    ltc, offc = _coords(d, cols, 1)
    dev = st.device
    st[ltr.to(dev)[:, None], ltc.to(dev)[None, :],
       offr.to(dev)[:, None], offc.to(dev)[None, :]] = block.to(st.dtype)


def owned_globals(dist, axis: int, limit: int) -> np.ndarray:
    """Sorted global element indices this rank owns along ``axis``."""
    if axis == 0:
        nt, b, grid, rank, src = (dist.nr_tiles[0], dist.mb, dist.grid_rows,
                                  dist.rank_row, dist.src_rank_row)
    else:
        nt, b, grid, rank, src = (dist.nr_tiles[1], dist.nb, dist.grid_cols,
                                  dist.rank_col, dist.src_rank_col)
    out = []
    for t in range(nt):
        if ix.rank_of_tile(t, grid, src) == rank:
            out.append(np.arange(t * b, min((t + 1) * b, limit)))
    return np.concatenate(out) if out else np.zeros(0, dtype=np.int64)


def owned_globals_of_rank(dist, axis: int, limit: int, rank: int
                          ) -> np.ndarray:
    if axis == 0:
        nt, b, grid, src = (dist.nr_tiles[0], dist.mb, dist.grid_rows,
                            dist.src_rank_row)
    else:
        nt, b, grid, src = (dist.nr_tiles[1], dist.nb, dist.grid_cols,
                            dist.src_rank_col)
    out = []
    for t in range(nt):
        if ix.rank_of_tile(t, grid, src) == rank:
            out.append(np.arange(t * b, min((t + 1) * b, limit)))
    return np.concatenate(out) if out else np.zeros(0, dtype=np.int64)
