"""Unit tests for 1D/2D block-cyclic index math.

Mirrors the reference's test/unit/matrix/test_distribution.cpp coverage:
conversions, ownership, local counts, degenerate shapes.
"""

import pytest

from dlaf_amd.core import index as ix
from dlaf_amd.core.distribution import Distribution


@pytest.mark.parametrize("size,tile", [(0, 4), (1, 4), (4, 4), (5, 4), (16, 4), (17, 5)])
def test_num_tiles(size, tile):
    nt = ix.num_tiles(size, tile)
    assert nt == (size + tile - 1) // tile
    if nt:
        assert ix.tile_size_of(nt - 1, size, tile) == size - (nt - 1) * tile


@pytest.mark.parametrize("grid", [1, 2, 3, 4])
@pytest.mark.parametrize("src", [0, 1])
def test_cyclic_roundtrip(grid, src):
    if src >= grid:
        pytest.skip("src rank out of range")
    ntiles = 13
    counts = [0] * grid
    for t in range(ntiles):
        r = ix.rank_of_tile(t, grid, src)
        lt = ix.local_tile_of_global(t, grid)
        assert ix.global_tile_of_local(lt, grid, r, src) == t
        counts[r] += 1
    for r in range(grid):
        assert counts[r] == ix.num_local_tiles(ntiles, grid, r, src)


def test_next_local_tile():
    grid, rank = 3, 1
    # rank 1 owns global tiles 1, 4, 7, ...
    assert ix.next_local_tile(0, grid, rank) == 0
    assert ix.next_local_tile(1, grid, rank) == 0
    assert ix.next_local_tile(2, grid, rank) == 1
    assert ix.next_local_tile(4, grid, rank) == 1
    assert ix.next_local_tile(5, grid, rank) == 2


@pytest.mark.parametrize(
    "m,n,mb,nb,gr,gc",
    [
        (0, 0, 4, 4, 1, 1),
        (10, 10, 3, 3, 1, 1),
        (10, 8, 3, 2, 2, 3),
        (33, 33, 8, 8, 2, 2),
        (64, 64, 16, 16, 3, 2),
        (5, 5, 8, 8, 2, 3),  # single (partial) tile
    ],
)
def test_distribution_consistency(m, n, mb, nb, gr, gc):
    total = 0
    seen = set()
    for rr in range(gr):
        for rc in range(gc):
            d = Distribution(m, n, mb, nb, gr, gc, rr, rc)
            lr, lc = d.local_nr_tiles
            lsz = d.local_size
            # local size consistency
            rows = sum(d.tile_size_of(d.global_tile_of_local((li, 0)))[0] for li in range(lr))
            cols = sum(d.tile_size_of(d.global_tile_of_local((0, lj)))[1] for lj in range(lc))
            if lr and lc:
                assert (rows, cols) == lsz
            for t in d.iter_local_tiles_global():
                assert d.rank_of_tile(t) == (rr, rc)
                assert t not in seen
                seen.add(t)
                total += 1
    d0 = Distribution(m, n, mb, nb, gr, gc, 0, 0)
    assert total == d0.nr_tiles[0] * d0.nr_tiles[1]


def test_element_tile_conversions():
    d = Distribution(20, 20, 6, 6, 2, 2, 0, 0)
    assert d.global_tile_of_element((0, 0)) == (0, 0)
    assert d.global_tile_of_element((5, 6)) == (0, 1)
    assert d.global_tile_of_element((19, 19)) == (3, 3)
    assert d.tile_size_of((3, 3)) == (2, 2)
    assert d.global_element_of_tile((2, 1)) == (12, 6)
