"""1D block-cyclic index arithmetic.

Pure integer math, the 1D building blocks composed into the 2D ``Distribution``.
Semantics match the reference's ``include/dlaf/util_distribution.h`` /
``include/dlaf/matrix/distribution.h`` conversion set (single tile per block):

* global element ``g`` lives in global tile ``g // tile``, at in-tile offset
  ``g % tile``;
* global tile ``t`` is owned by rank ``(t + src) % grid``;
* on its owner, global tile ``t`` is local tile ``t // grid``.
"""

from __future__ import annotations


def num_tiles(size: int, tile: int) -> int:
    """Number of tiles covering ``size`` elements (last tile may be partial)."""
    return (size + tile - 1) // tile if size > 0 else 0


def tile_of_element(g: int, tile: int) -> int:
    return g // tile


def tile_element(g: int, tile: int) -> int:
    return g % tile


def tile_size_of(t: int, size: int, tile: int) -> int:
    """Extent of global tile ``t`` (the trailing tile may be partial)."""
    return min(tile, size - t * tile)


def rank_of_tile(t: int, grid: int, src: int = 0) -> int:
    return (t + src) % grid


def local_tile_of_global(t: int, grid: int) -> int:
    """Local tile index of global tile ``t`` on its owning rank."""
    return t // grid


def global_tile_of_local(lt: int, grid: int, rank: int, src: int = 0) -> int:
    """Global tile index of local tile ``lt`` on ``rank``."""
    return lt * grid + (rank - src) % grid


def num_local_tiles(ntiles: int, grid: int, rank: int, src: int = 0) -> int:
    """Number of global tiles owned by ``rank``."""
    r = (rank - src) % grid
    return (ntiles - r + grid - 1) // grid if ntiles > r else 0


def next_local_tile(t: int, grid: int, rank: int, src: int = 0) -> int:
    """Smallest local tile index whose global tile is >= ``t`` on ``rank``.

    Mirrors ``Distribution::next_local_tile_from_global_tile`` — the canonical way
    algorithm loops restrict a global tile range to the calling rank.
    """
    r = (rank - src) % grid
    if t <= r:
        return 0
    return (t - r + grid - 1) // grid


def local_size(size: int, tile: int, grid: int, rank: int, src: int = 0) -> int:
    """Number of elements owned by ``rank`` in a 1D block-cyclic layout."""
    nt = num_tiles(size, tile)
    lnt = num_local_tiles(nt, grid, rank, src)
    if lnt == 0:
        return 0
    last_gt = global_tile_of_local(lnt - 1, grid, rank, src)
    return (lnt - 1) * tile + tile_size_of(last_gt, size, tile)
