"""TRTRI / POTRI correctness vs torch reference.

Mirrors ``test/unit/inverse/test_{triangular_inverse,
inverse_from_cholesky_factor}.cpp``.
"""

import pytest
import torch

from dlaf_amd import (
    Matrix, CommGrid, UpLo, Diag,
    triangular_inverse, inverse_from_cholesky_factor,
)
from dlaf_amd.matrix import util as mutil

from dist_utils import run_distributed


def _tri(a, unit):
    t = torch.tril(a)
    if unit:
        n = a.shape[0]
        t = torch.tril(a, -1) + torch.eye(n, dtype=a.dtype)
    return t


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("diag", [Diag.NonUnit, Diag.Unit])
@pytest.mark.parametrize("n,nb", [(4, 4), (16, 4), (21, 5)])
def test_trtri_local_cpu(dtype, diag, n, nb):
    mat = Matrix.create(n, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian_positive_definite(mat, seed=5)
    a = mat.to_global()
    triangular_inverse(UpLo.Lower, diag, mat)
    unit = diag == Diag.Unit
    want = torch.linalg.inv(_tri(a, unit))
    got = torch.tril(mat.to_global(), -1 if unit else 0)
    want_cmp = torch.tril(want, -1 if unit else 0)
    err = (got - want_cmp).abs().max().item()
    assert err < 1e-10 * n, f"err={err}"
    if unit:  # diagonal must be untouched
        dg = (mat.to_global().diagonal() - a.diagonal()).abs().max().item()
        assert dg == 0.0


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("n,nb", [(16, 4), (21, 5)])
def test_potri_local_cpu(dtype, n, nb):
    mat = Matrix.create(n, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian_positive_definite(mat, seed=6)
    a = mat.to_global()
    L = torch.linalg.cholesky(a)
    mat.set_from_global(L)
    inverse_from_cholesky_factor(UpLo.Lower, mat)
    want = torch.linalg.inv(a)
    err = (torch.tril(mat.to_global()) - torch.tril(want)).abs().max().item()
    assert err < 1e-9 * n, f"err={err}"


def _dist_worker(rank, ws, gr, gc, which, n, nb, dtype_str):
    dtype = getattr(torch, dtype_str)
    grid = CommGrid(gr, gc)
    mat = Matrix.create(n, n, nb, nb, dtype=dtype, grid=grid)
    mutil.set_random_hermitian_positive_definite(mat, seed=8)
    a = mat.to_global()
    if which == "trtri":
        triangular_inverse(UpLo.Lower, Diag.NonUnit, mat, grid)
        want = torch.linalg.inv(torch.tril(a))
    else:
        L = torch.linalg.cholesky(a)
        mat.set_from_global(L)
        inverse_from_cholesky_factor(UpLo.Lower, mat, grid)
        want = torch.linalg.inv(a)
    got = torch.tril(mat.to_global())
    return (got - torch.tril(want)).abs().max().item()


@pytest.mark.parametrize("which", ["trtri", "potri"])
@pytest.mark.parametrize("gr,gc", [(2, 2), (1, 2)])
def test_inverse_dist_cpu(which, gr, gc):
    errs = run_distributed(_dist_worker, gr * gc, args=(gr, gc, which, 24, 4, "float64"))
    for e in errs:
        assert e < 1e-9, f"err={e}"


@pytest.mark.parametrize("which", ["trtri", "potri"])
def test_inverse_dist_cpu_complex(which):
    errs = run_distributed(_dist_worker, 2, args=(2, 1, which, 18, 5, "complex128"))
    for e in errs:
        assert e < 1e-9, f"err={e}"


def test_trtri_upper_local():
    n, nb = 18, 6
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64)
    mutil.set_random_hermitian_positive_definite(mat, seed=9)
    a = mat.to_global()
    triangular_inverse(UpLo.Upper, Diag.NonUnit, mat)
    want = torch.linalg.inv(torch.triu(a))
    err = (torch.triu(mat.to_global()) - torch.triu(want)).abs().max().item()
    assert err < 1e-10 * n, f"err={err}"


def test_potri_upper_local():
    n, nb = 16, 4
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64)
    mutil.set_random_hermitian_positive_definite(mat, seed=10)
    a = mat.to_global()
    U = torch.linalg.cholesky(a).mH
    mat.set_from_global(torch.triu(U))
    inverse_from_cholesky_factor(UpLo.Upper, mat)
    want = torch.linalg.inv(a)
    err = (torch.triu(mat.to_global()) - torch.triu(want)).abs().max().item()
    assert err < 1e-9 * n, f"err={err}"
