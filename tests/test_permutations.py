"""Permutation tests (reference test/unit/permutations)."""
import torch
from dlaf_amd import Matrix, CommGrid
from dlaf_amd.matrix import util as mutil
from dlaf_amd.algs.permutations import permute_columns, permute_rows
from dist_utils import run_distributed


def test_permute_local():
    m = Matrix.create(12, 12, 4, 4)
    mutil.set_random(m, seed=1)
    d = Matrix.create(12, 12, 4, 4)
    g = torch.randperm(12, generator=torch.Generator().manual_seed(2))
    a = m.to_global()
    permute_columns(m, g, d)
    assert torch.equal(d.to_global(), a[:, g])
    permute_rows(m, g, d)
    assert torch.equal(d.to_global(), a[g, :])


def _worker(rank, ws, gr, gc):
    grid = CommGrid(gr, gc)
    m = Matrix.create(16, 16, 4, 4, grid=grid)
    mutil.set_random(m, seed=1)
    d = Matrix.create(16, 16, 4, 4, grid=grid)
    g = torch.randperm(16, generator=torch.Generator().manual_seed(3))
    a = m.to_global()
    permute_columns(m, g, d, grid)
    return (d.to_global() - a[:, g]).abs().max().item()


def test_permute_dist():
    for e in run_distributed(_worker, 4, args=(2, 2)):
        assert e == 0.0


def _worker_rows(rank, ws, gr, gc):
    from dlaf_amd.algs.permutations import permute_rows
    grid = CommGrid(gr, gc)
    m = Matrix.create(20, 12, 4, 4, grid=grid)
    mutil.set_random(m, seed=7)
    d = Matrix.create(20, 12, 4, 4, grid=grid)
    g = torch.randperm(20, generator=torch.Generator().manual_seed(9))
    a = m.to_global()
    permute_rows(m, g, d, grid)
    return (d.to_global() - a[g, :]).abs().max().item()


def test_permute_rows_dist_6rank():
    for e in run_distributed(_worker_rows, 6, args=(2, 3)):
        assert e == 0.0


def test_permute_cols_dist_6rank():
    for e in run_distributed(_worker, 6, args=(3, 2)):
        assert e == 0.0
