"""Cholesky correctness: local CPU, distributed CPU (gloo), vs torch reference.

Mirrors the reference's test/unit/factorization/test_cholesky.cpp strategy:
typed over dtypes x grids x {size, block} tables including degenerate and
non-divisible shapes; verification against an independent reference factor.
"""

import pytest
import torch

from dlaf_amd import Matrix, CommGrid, UpLo, cholesky_factorization
from dlaf_amd.matrix import util as mutil

from dist_utils import run_distributed

DTYPES = [torch.float32, torch.float64, torch.complex64, torch.complex128]
SIZES = [(0, 4), (4, 4), (8, 4), (24, 8), (33, 8), (25, 6)]


def _tol(dtype):
    return 5e-5 if dtype in (torch.float32, torch.complex64) else 1e-11


def _check_factor(mat, a_ref):
    """Compare tril of the computed factor with torch.linalg.cholesky."""
    n = a_ref.shape[0]
    if n == 0:
        return
    got = torch.tril(mat.to_global())
    want = torch.linalg.cholesky(a_ref)
    err = (got - want).abs().max().item()
    scale = max(1.0, want.abs().max().item())
    assert err <= _tol(mat.dtype) * scale * max(1, n), f"err={err}"


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("n,nb", SIZES)
def test_cholesky_local_cpu(dtype, n, nb):
    mat = Matrix.create(n, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian_positive_definite(mat, seed=7)
    a_ref = mat.to_global()
    cholesky_factorization(UpLo.Lower, mat)
    _check_factor(mat, a_ref)


def _dist_cholesky_worker(rank, world_size, gr, gc, n, nb, dtype_str):
    dtype = getattr(torch, dtype_str)
    grid = CommGrid(gr, gc)
    mat = Matrix.create(n, n, nb, nb, dtype=dtype, grid=grid)
    mutil.set_random_hermitian_positive_definite(mat, seed=11)
    a_ref = mat.to_global()
    cholesky_factorization(UpLo.Lower, mat)
    got = torch.tril(mat.to_global())
    want = torch.linalg.cholesky(a_ref)
    err = (got - want).abs().max().item()
    return err


@pytest.mark.parametrize("gr,gc", [(1, 2), (2, 1), (2, 2)])
@pytest.mark.parametrize("n,nb", [(24, 4), (33, 8)])
def test_cholesky_dist_cpu(gr, gc, n, nb):
    ws = gr * gc
    errs = run_distributed(_dist_cholesky_worker, ws, args=(gr, gc, n, nb, "float64"))
    for e in errs:
        assert e < 1e-11 * n, f"err={e}"


@pytest.mark.parametrize("dtype_str", ["complex128", "float32"])
def test_cholesky_dist_cpu_dtypes(dtype_str):
    errs = run_distributed(_dist_cholesky_worker, 2, args=(1, 2, 24, 6, dtype_str))
    tol = 1e-3 if dtype_str == "float32" else 1e-10
    for e in errs:
        assert e < tol, f"err={e}"


def test_cholesky_upper_local():
    n, nb = 24, 8
    mat = Matrix.create(n, n, nb, nb, dtype=torch.complex128)
    mutil.set_random_hermitian_positive_definite(mat, seed=7)
    a_ref = mat.to_global()
    a_ref = torch.tril(a_ref) + torch.tril(a_ref, -1).mH
    cholesky_factorization(UpLo.Upper, mat)
    U = torch.triu(mat.to_global())
    err = (U.mH @ U - a_ref).abs().max().item()
    assert err < 1e-11 * n, f"err={err}"


def test_cholesky_complex64_local():
    n, nb = 24, 8
    mat = Matrix.create(n, n, nb, nb, dtype=torch.complex64)
    mutil.set_random_hermitian_positive_definite(mat, seed=3)
    a_ref = mat.to_global().to(torch.complex128)
    cholesky_factorization(UpLo.Lower, mat)
    got = torch.tril(mat.to_global()).to(torch.complex128)
    want = torch.linalg.cholesky(a_ref)
    assert (got - want).abs().max().item() < 1e-3


@pytest.mark.timeout(600)
def test_cholesky_dist_cpu_2x4():
    """The driver's 8-GPU SCALE grid shape (2x4), 8 gloo ranks."""
    errs = run_distributed(_dist_cholesky_worker, 8, args=(2, 4, 40, 8, "float64"))
    for e in errs:
        assert e < 1e-11 * 40, f"err={e}"


def test_cholesky_upper_native_local():
    """Native Upper path (no storage transpose): A = U^H U on upper tiles."""
    import torch as _t
    for n, nb in ((96, 32), (130, 48)):
        mat = Matrix.create(n, n, nb, nb, dtype=_t.float64, device="cpu")
        mutil.set_random_hermitian_positive_definite(mat, seed=9)
        A = mat.to_global()
        A = _t.triu(A.mT) + _t.triu(A.mT, 1).mT  # hermitianize via upper
        mat.set_from_global(A)
        cholesky_factorization(UpLo.Upper, mat)
        U = _t.triu(mat.to_global())
        err = (U.mH @ U - A).abs().max().item()
        assert err < 1e-10 * n, err


def test_cholesky_upper_native_complex():
    import torch as _t
    n, nb = 96, 32
    mat = Matrix.create(n, n, nb, nb, dtype=_t.complex128, device="cpu")
    mutil.set_random_hermitian_positive_definite(mat, seed=3)
    A = mat.to_global()
    A = _t.tril(A) + _t.tril(A, -1).mH
    mat.set_from_global(A.clone())
    cholesky_factorization(UpLo.Upper, mat)
    U = _t.triu(mat.to_global())
    err = (U.mH @ U - A).abs().max().item()
    assert err < 1e-10 * n, err
