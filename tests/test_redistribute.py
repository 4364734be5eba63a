"""Redistribution tests (reference ``test/unit/matrix/test_copy.cpp`` /
miniapp_redistribution): copy between matrices with different block sizes
and grids, local and multi-rank."""
import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from dist_utils import run_distributed
from dlaf_amd import Matrix, CommGrid
from dlaf_amd.matrix import util as mutil
from dlaf_amd.algs.redistribute import redistribute


@pytest.mark.parametrize("nb_src,nb_dst", [(8, 16), (16, 8), (12, 20)])
def test_redistribute_local(nb_src, nb_dst):
    m, n = 52, 36
    src = Matrix.create(m, n, nb_src, nb_src, dtype=torch.float64)
    mutil.set_random(src, seed=2)
    dst = Matrix.create(m, n, nb_dst, nb_dst, dtype=torch.float64)
    redistribute(src, dst)
    assert torch.equal(src.to_global(), dst.to_global())


def _worker(rank, ws):
    grid = CommGrid(1, 2)
    m, n = 48, 40
    src = Matrix.create(m, n, 8, 8, dtype=torch.float64, grid=grid)
    mutil.set_random(src, seed=3)
    dst = Matrix.create(m, n, 16, 16, dtype=torch.float64, grid=grid)
    redistribute(src, dst)
    return (src.to_global() - dst.to_global()).abs().max().item()


def test_redistribute_dist():
    errs = run_distributed(_worker, 2)
    for e in errs:
        assert e == 0.0


def _worker4(rank, ws):
    grid = CommGrid(2, 2)
    m, n = 52, 36
    src = Matrix.create(m, n, 8, 8, dtype=torch.complex128, grid=grid)
    mutil.set_random(src, seed=5)
    dst = Matrix.create(m, n, 12, 12, dtype=torch.complex128, grid=grid)
    redistribute(src, dst)
    return (src.to_global() - dst.to_global()).abs().max().item()


def test_redistribute_dist_2x2_complex():
    for e in run_distributed(_worker4, 4):
        assert e == 0.0
