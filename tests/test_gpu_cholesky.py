"""End-to-end GPU Cholesky vs torch CPU reference."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from dlaf_amd import Matrix, UpLo, cholesky_factorization  # noqa: E402
from dlaf_amd.matrix import util as mutil  # noqa: E402


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128, torch.float32])
@pytest.mark.parametrize("n,nb", [(512, 128), (1024, 256), (1000, 256), (768, 512)])
def test_cholesky_local_gpu(dtype, n, nb):
    mat = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=7)
    a_ref = mat.to_global().cpu().to(torch.complex128 if dtype.is_complex else torch.float64)
    cholesky_factorization(UpLo.Lower, mat)
    torch.cuda.synchronize()
    got = torch.tril(mat.to_global().cpu().to(a_ref.dtype))
    want = torch.linalg.cholesky(a_ref)
    err = (got - want).abs().max().item()
    scale = want.abs().max().item()
    tol = 1e-3 if dtype in (torch.float32,) else 1e-9
    assert err <= tol * scale * n, f"err={err} scale={scale}"


def test_cholesky_residual_gpu():
    """||A - L L^H|| / ||A|| residual check, the miniapp verification style."""
    n, nb = 2048, 512
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=1)
    a = mat.to_global()
    cholesky_factorization(UpLo.Lower, mat)
    torch.cuda.synchronize()
    L = torch.tril(mat.to_global())
    res = (a - L @ L.mH).abs().max().item() / a.abs().max().item()
    assert res < 1e-13 * n, f"residual={res}"


def test_cholesky_dist_gpu_path_single_rank():
    """Exercise the lookahead distributed-GPU code path (streams, events,
    double-buffered panels, plan tables) on a trivial 1x1 grid — the RCCL
    collectives are no-ops but every other statement runs."""
    from dlaf_amd.algs.cholesky import _cholesky_dist_gpu
    from dlaf_amd import CommGrid
    n, nb = 1536, 256
    grid = CommGrid(1, 1)
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=5)
    a = mat.to_global().cpu()
    _cholesky_dist_gpu(mat, grid)
    torch.cuda.synchronize()
    got = torch.tril(mat.to_global().cpu())
    want = torch.linalg.cholesky(a)
    err = (got - want).abs().max().item()
    assert err < 1e-10 * n, f"err={err}"


@pytest.mark.timeout(300)
def test_cholesky_trsm_complex64_gpu():
    """complex64 coverage of the GPU factor + solve path."""
    from dlaf_amd import Side, Op, Diag, triangular_solver
    n, nb = 1024, 256
    A = Matrix.create(n, n, nb, nb, dtype=torch.complex64, device="cuda")
    mutil.set_random_hermitian_positive_definite(A, seed=9)
    a0 = A.to_global().cpu()
    cholesky_factorization(UpLo.Lower, A)
    torch.cuda.synchronize()
    L = torch.tril(A.to_global().cpu())
    err = (L @ L.mH - a0).abs().max().item() / a0.abs().max().item()
    assert err < 1e-4, f"chol err={err}"
    B = Matrix.create(n, 512, nb, nb, dtype=torch.complex64, device="cuda")
    mutil.set_random(B, seed=10)
    b0 = B.to_global().cpu()
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0, A, B)
    torch.cuda.synchronize()
    want = torch.linalg.solve(L.to(torch.complex128), b0.to(torch.complex128))
    err = (B.to_global().cpu().to(torch.complex128) - want).abs().max().item()
    assert err < 1e-2, f"trsm err={err}"
