"""HEGST correctness vs dense reference: A' = inv(L) herm(A) inv(L)^H.

Mirrors ``test/unit/eigensolver/test_gen_to_std.cpp``.
"""

import pytest
import torch

from dlaf_amd import Matrix, CommGrid, UpLo, generalized_to_standard
from dlaf_amd.matrix import util as mutil

from dist_utils import run_distributed


def _herm(a):
    return torch.tril(a) + torch.tril(a, -1).mH


def _run(n, nb, dtype, grid=None):
    A = Matrix.create(n, n, nb, nb, dtype=dtype, grid=grid)
    L = Matrix.create(n, n, nb, nb, dtype=dtype, grid=grid)
    mutil.set_random_hermitian(A, seed=21)
    mutil.set_random_hermitian_positive_definite(L, seed=22)
    b = L.to_global()
    Lf = torch.linalg.cholesky(b)
    L.set_from_global(Lf)
    a0 = A.to_global()
    generalized_to_standard(UpLo.Lower, A, L, grid)
    linv = torch.linalg.inv(Lf)
    want = linv @ _herm(a0) @ linv.mH
    got = _herm(A.to_global())
    return (got - want).abs().max().item()


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("n,nb", [(4, 4), (16, 4), (21, 5)])
def test_hegst_local_cpu(dtype, n, nb):
    err = _run(n, nb, dtype)
    assert err < 1e-9 * n, f"err={err}"


def _dist_worker(rank, ws, gr, gc, n, nb, dtype_str):
    return _run(n, nb, getattr(torch, dtype_str), grid=CommGrid(gr, gc))


@pytest.mark.parametrize("gr,gc", [(2, 2), (1, 2)])
def test_hegst_dist_cpu(gr, gc):
    errs = run_distributed(_dist_worker, gr * gc, args=(gr, gc, 24, 4, "float64"))
    for e in errs:
        assert e < 1e-9, f"err={e}"


def test_hegst_dist_cpu_complex():
    errs = run_distributed(_dist_worker, 2, args=(2, 1, 18, 5, "complex128"))
    for e in errs:
        assert e < 1e-9, f"err={e}"


def test_hegst_upper_local():
    n, nb = 16, 4
    dtype = torch.complex128
    A = Matrix.create(n, n, nb, nb, dtype=dtype)
    L = Matrix.create(n, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian(A, seed=31)
    mutil.set_random_hermitian_positive_definite(L, seed=32)
    b = L.to_global()
    b = torch.tril(b) + torch.tril(b, -1).mH
    U = torch.linalg.cholesky(b).mH
    L.set_from_global(torch.triu(U))
    a0 = A.to_global()
    a0h = torch.triu(a0) + torch.triu(a0, 1).mH
    A.set_from_global(torch.triu(a0))
    generalized_to_standard(UpLo.Upper, A, L)
    uinv = torch.linalg.inv(U)
    want = uinv.mH @ a0h @ uinv
    got = A.to_global()
    got = torch.triu(got) + torch.triu(got, 1).mH
    err = (got - want).abs().max().item()
    assert err < 1e-9 * n, f"err={err}"


def test_hegst_upper_native_local():
    """Native Upper HEGST (no storage transposes) vs dense reference."""
    import torch
    from dlaf_amd import Matrix, UpLo, generalized_to_standard
    from dlaf_amd.matrix import util as mutil
    for dtype in (torch.float64, torch.complex128):
        for n, nb in ((96, 32), (130, 48)):
            a = Matrix.create(n, n, nb, nb, dtype=dtype)
            mutil.set_random_hermitian(a, seed=3)
            A = a.to_global()
            A = torch.tril(A) + torch.tril(A, -1).mH
            a.set_from_global(A.clone())
            u = Matrix.create(n, n, nb, nb, dtype=dtype)
            mutil.set_random_hermitian_positive_definite(u, seed=4)
            B = u.to_global()
            B = torch.tril(B) + torch.tril(B, -1).mH
            U = torch.linalg.cholesky(B, upper=True)
            u.set_from_global(U.clone())
            generalized_to_standard(UpLo.Upper, a, u)
            got = a.to_global()
            got = torch.triu(got) + torch.triu(got, 1).mH
            Ui = torch.linalg.solve_triangular(
                U, torch.eye(n, dtype=dtype), upper=True)
            want = Ui.mH @ A @ Ui
            err = (got - want).abs().max().item()
            assert err < 1e-11 * n, (dtype, n, nb, err)
