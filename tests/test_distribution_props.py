"""Property-based Distribution tests (hypothesis): the reference's
test_distribution/test_util_distribution exercise the index-math conversion
table exhaustively; here random (size, block, grid, rank, src) configurations
must satisfy the roundtrip and partition invariants."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from dlaf_amd.core.distribution import Distribution


@st.composite
def dists(draw):
    m = draw(st.integers(1, 200))
    n = draw(st.integers(1, 200))
    mb = draw(st.integers(1, 64))
    nb = draw(st.integers(1, 64))
    gr = draw(st.integers(1, 4))
    gc = draw(st.integers(1, 4))
    rr = draw(st.integers(0, gr - 1))
    rc = draw(st.integers(0, gc - 1))
    sr = draw(st.integers(0, gr - 1))
    sc = draw(st.integers(0, gc - 1))
    return Distribution(m, n, mb, nb, gr, gc, rr, rc, sr, sc)


@given(dists())
@settings(max_examples=200, deadline=None)
def test_local_global_tile_roundtrip(d):
    nrt = d.nr_tiles
    for gi in range(0, nrt[0], max(1, nrt[0] // 5)):
        for gj in range(0, nrt[1], max(1, nrt[1] // 5)):
            owner = d.rank_of_tile((gi, gj))
            if owner == (d.rank_row, d.rank_col):
                li, lj = d.local_tile_of_global((gi, gj))
                assert d.global_tile_of_local((li, lj)) == (gi, gj)
                lr, lc = d.local_nr_tiles
                assert 0 <= li < lr and 0 <= lj < lc


@given(dists())
@settings(max_examples=200, deadline=None)
def test_tile_partition_covers_matrix(d):
    """Sum of local tile element counts over all ranks == m*n."""
    total = 0
    for rr in range(d.grid_rows):
        for rc in range(d.grid_cols):
            dd = Distribution(d.m, d.n, d.mb, d.nb, d.grid_rows, d.grid_cols,
                              rr, rc, d.src_rank_row, d.src_rank_col)
            lr, lc = dd.local_nr_tiles
            for li in range(lr):
                for lj in range(lc):
                    gi, gj = dd.global_tile_of_local((li, lj))
                    tr, tc = dd.tile_size_of((gi, gj))
                    total += tr * tc
    assert total == d.m * d.n


@given(dists())
@settings(max_examples=100, deadline=None)
def test_next_local_tile_monotone(d):
    lr = d.local_nr_tiles[0]
    prev = 0
    for k in range(d.nr_tiles[0] + 1):
        v = d.next_local_tile_row(k)
        assert prev <= v <= lr
        prev = v
