"""8-rank (2x4 — the driver's 8-GPU SCALE shape) gloo tests of every
distributed algorithm (round-1 verdict item 4: the multi-GPU path must be
covered by multi-process CPU tests at the production grid shape)."""

import numpy as np
import pytest
import torch

from dlaf_amd import (Matrix, CommGrid, UpLo, Op, Side, Diag,
                      cholesky_factorization, triangular_solver,
                      triangular_multiplication, hermitian_multiplication,
                      triangular_inverse, inverse_from_cholesky_factor,
                      generalized_to_standard)
from dlaf_amd.matrix import util as mutil

from dist_utils import run_distributed

pytestmark = pytest.mark.timeout(900)

N, NB = 40, 8


def _mk(n, nb, grid, dtype=torch.float64, seed=1, spd=False, general=None):
    m = Matrix.create(n, general if general else n, nb, nb, dtype=dtype,
                      grid=grid)
    if spd:
        mutil.set_random_hermitian_positive_definite(m, seed=seed)
    else:
        mutil.set_random(m, seed=seed)
    return m


def _herm(a):
    return torch.tril(a) + torch.tril(a, -1).mH


def _w_trsm(rank, ws):
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=1, spd=True)
    b = _mk(N, NB, grid, seed=2)
    A = torch.tril(a.to_global())
    B = b.to_global().clone()
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit,
                      1.0, a, b, grid)
    X = b.to_global()
    return (A @ X - B).abs().max().item()


def _w_trmm(rank, ws):
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=3, spd=True)
    b = _mk(N, NB, grid, seed=4)
    A = torch.tril(a.to_global())
    B = b.to_global().clone()
    triangular_multiplication(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit,
                              1.0, a, b, grid)
    return (b.to_global() - A @ B).abs().max().item()


def _w_hemm(rank, ws):
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=5, spd=True)
    b = _mk(N, NB, grid, seed=6)
    c = _mk(N, NB, grid, seed=7)
    A = _herm(a.to_global())
    B = b.to_global().clone()
    C = c.to_global().clone()
    hermitian_multiplication(Side.Left, UpLo.Lower, 1.5, a, b, 0.5, c, grid)
    return (c.to_global() - (1.5 * A @ B + 0.5 * C)).abs().max().item()


def _w_trtri(rank, ws):
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=8, spd=True)
    A = torch.tril(a.to_global())
    triangular_inverse(UpLo.Lower, Diag.NonUnit, a, grid)
    X = torch.tril(a.to_global())
    eye = torch.eye(N, dtype=A.dtype)
    return (A @ X - eye).abs().max().item()


def _w_potri(rank, ws):
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=9, spd=True)
    A = _herm(a.to_global())
    cholesky_factorization(UpLo.Lower, a, grid)
    inverse_from_cholesky_factor(UpLo.Lower, a, grid)
    Ainv = _herm(a.to_global())
    eye = torch.eye(N, dtype=A.dtype)
    return (A @ Ainv - eye).abs().max().item()


def _w_hegst(rank, ws):
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=10, spd=True)
    bm = _mk(N, NB, grid, seed=11, spd=True)
    A = _herm(a.to_global())
    cholesky_factorization(UpLo.Lower, bm, grid)
    L = torch.tril(bm.to_global())
    generalized_to_standard(UpLo.Lower, a, bm, grid)
    got = _herm(a.to_global())
    Li = torch.linalg.solve_triangular(L, torch.eye(N, dtype=L.dtype),
                                       upper=False)
    want = Li @ A @ Li.mH
    return (got - want).abs().max().item()


def _w_norm(rank, ws):
    from dlaf_amd.algs.norm import max_norm
    grid = CommGrid(2, 4)
    a = _mk(N, NB, grid, seed=12)
    v = max_norm(a, grid=grid)
    want = a.to_global().abs().max().item()
    return abs(float(v) - want)


@pytest.mark.parametrize("worker", [
    _w_trsm, _w_trmm, _w_hemm, _w_trtri, _w_potri, _w_hegst, _w_norm,
], ids=["trsm", "trmm", "hemm", "trtri", "potri", "hegst", "norm"])
def test_dist_8rank_2x4(worker):
    for e in run_distributed(worker, 8):
        assert e < 1e-10 * N, e


def _w_gen_eig(rank, ws):
    from dlaf_amd import hermitian_generalized_eigensolver
    grid = CommGrid(2, 4)
    n, nb, band = 64, 8, 4
    a = _mk(n, nb, grid, seed=13)
    bm = _mk(n, nb, grid, seed=14, spd=True)
    A = _herm(a.to_global())
    a.set_from_global(A.clone())
    B = _herm(bm.to_global())
    bm.set_from_global(B.clone())
    w, evecs = hermitian_generalized_eigensolver(UpLo.Lower, a, bm, grid,
                                                 band=band)
    E = evecs.to_global()
    R = A @ E - B @ E @ torch.diag(w.to(E.dtype))
    return float(R.abs().max())


def test_dist_8rank_gen_eigensolver():
    for e in run_distributed(_w_gen_eig, 8):
        assert e < 1e-9, e
