"""2D block-cyclic distribution of an m x n matrix over an r x c process grid.

Counterpart of the reference's ``include/dlaf/matrix/distribution.h`` (the full
global<->local element/tile conversion set, rank-of-tile, local tile counts), with
the reference's "single tile per block" restriction made structural: block == tile.
Tile indices are (row, col) pairs; ranks are (row, col) pairs in the grid.

Everything here is pure integer math and is identical on every rank — no torch, no
communication.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Iterator, Tuple

from . import index as ix

Coord = Tuple[int, int]


@dataclass(frozen=True)
class Distribution:
    m: int
    n: int
    mb: int
    nb: int
    grid_rows: int = 1
    grid_cols: int = 1
    rank_row: int = 0
    rank_col: int = 0
    src_rank_row: int = 0
    src_rank_col: int = 0

    def __post_init__(self):
        assert self.m >= 0 and self.n >= 0, (self.m, self.n)
        assert self.mb > 0 and self.nb > 0, (self.mb, self.nb)
        assert 0 <= self.rank_row < self.grid_rows
        assert 0 <= self.rank_col < self.grid_cols
        assert 0 <= self.src_rank_row < self.grid_rows
        assert 0 <= self.src_rank_col < self.grid_cols

    # ---- global sizes ----
    @property
    def size(self) -> Coord:
        return (self.m, self.n)

    @property
    def tile_size(self) -> Coord:
        return (self.mb, self.nb)

    @property
    def nr_tiles(self) -> Coord:
        return (ix.num_tiles(self.m, self.mb), ix.num_tiles(self.n, self.nb))

    @property
    def is_local(self) -> bool:
        return self.grid_rows == 1 and self.grid_cols == 1

    @property
    def square_tiles(self) -> bool:
        return self.mb == self.nb

    # ---- tile geometry ----
    def tile_size_of(self, tile: Coord) -> Coord:
        """Actual (rows, cols) extent of global tile (i, j)."""
        i, j = tile
        return (ix.tile_size_of(i, self.m, self.mb), ix.tile_size_of(j, self.n, self.nb))

    def global_element_of_tile(self, tile: Coord) -> Coord:
        return (tile[0] * self.mb, tile[1] * self.nb)

    def global_tile_of_element(self, el: Coord) -> Coord:
        return (el[0] // self.mb, el[1] // self.nb)

    # ---- ownership ----
    def rank_of_tile(self, tile: Coord) -> Coord:
        return (
            ix.rank_of_tile(tile[0], self.grid_rows, self.src_rank_row),
            ix.rank_of_tile(tile[1], self.grid_cols, self.src_rank_col),
        )

    def is_tile_local(self, tile: Coord) -> bool:
        return self.rank_of_tile(tile) == (self.rank_row, self.rank_col)

    def rank_of_tile_row(self, i: int) -> int:
        return ix.rank_of_tile(i, self.grid_rows, self.src_rank_row)

    def rank_of_tile_col(self, j: int) -> int:
        return ix.rank_of_tile(j, self.grid_cols, self.src_rank_col)

    # ---- global <-> local tiles ----
    def local_tile_of_global(self, tile: Coord) -> Coord:
        assert self.is_tile_local(tile), (tile, self.rank_of_tile(tile))
        return (
            ix.local_tile_of_global(tile[0], self.grid_rows),
            ix.local_tile_of_global(tile[1], self.grid_cols),
        )

    def global_tile_of_local(self, ltile: Coord) -> Coord:
        return (
            ix.global_tile_of_local(ltile[0], self.grid_rows, self.rank_row, self.src_rank_row),
            ix.global_tile_of_local(ltile[1], self.grid_cols, self.rank_col, self.src_rank_col),
        )

    @property
    def local_nr_tiles(self) -> Coord:
        nt_r, nt_c = self.nr_tiles
        return (
            ix.num_local_tiles(nt_r, self.grid_rows, self.rank_row, self.src_rank_row),
            ix.num_local_tiles(nt_c, self.grid_cols, self.rank_col, self.src_rank_col),
        )

    @property
    def local_size(self) -> Coord:
        return (
            ix.local_size(self.m, self.mb, self.grid_rows, self.rank_row, self.src_rank_row),
            ix.local_size(self.n, self.nb, self.grid_cols, self.rank_col, self.src_rank_col),
        )

    def next_local_tile_row(self, i: int) -> int:
        """First local tile-row index whose global tile-row is >= i."""
        return ix.next_local_tile(i, self.grid_rows, self.rank_row, self.src_rank_row)

    def next_local_tile_col(self, j: int) -> int:
        return ix.next_local_tile(j, self.grid_cols, self.rank_col, self.src_rank_col)

    # ---- iteration ----
    def iter_local_tiles(self) -> Iterator[Coord]:
        """All local tiles in (col-major over local indices) order, as LOCAL coords."""
        lr, lc = self.local_nr_tiles
        for lj in range(lc):
            for li in range(lr):
                yield (li, lj)

    def iter_local_tiles_global(self) -> Iterator[Coord]:
        """All local tiles, as GLOBAL coords, col-major over local indices."""
        for li, lj in self.iter_local_tiles():
            yield self.global_tile_of_local((li, lj))

    # ---- derived distributions ----
    def for_rank(self, rank_row: int, rank_col: int) -> "Distribution":
        return Distribution(
            self.m, self.n, self.mb, self.nb,
            self.grid_rows, self.grid_cols, rank_row, rank_col,
            self.src_rank_row, self.src_rank_col,
        )

    def local_distribution(self) -> "Distribution":
        """The same matrix viewed as a non-distributed (1x1 grid) matrix."""
        return Distribution(self.m, self.n, self.mb, self.nb)
