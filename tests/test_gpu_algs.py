"""GPU correctness of the BLAS-3 algorithm suite + eigensolver pipeline.

Every algorithm runs on cuda with the native fused kernels and is compared
against a CPU torch fp64 reference (the project's numerics-test contract).
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from dlaf_amd import (  # noqa: E402
    Matrix, Side, UpLo, Op, Diag,
    triangular_solver, triangular_multiplication,
    hermitian_multiplication, general_multiplication,
    triangular_inverse, inverse_from_cholesky_factor,
    generalized_to_standard, max_norm,
    hermitian_eigensolver, hermitian_generalized_eigensolver,
)
from dlaf_amd.matrix import util as mutil  # noqa: E402


def _herm(a):
    return torch.tril(a) + torch.tril(a, -1).mH


def _t(x, op):
    return x if op is Op.NoTrans else (x.mT if op is Op.Trans else x.mH)


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("side,uplo,op", [
    (Side.Left, UpLo.Lower, Op.NoTrans), (Side.Left, UpLo.Upper, Op.ConjTrans),
    (Side.Right, UpLo.Lower, Op.ConjTrans), (Side.Right, UpLo.Upper, Op.NoTrans),
])
def test_trsm_gpu(dtype, side, uplo, op):
    m, n, nb = 1024, 768, 256
    k = m if side == Side.Left else n
    A = Matrix.create(k, k, nb, nb, dtype=dtype, device="cuda")
    B = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(A, seed=3)
    mutil.set_random(B, seed=4)
    a, b0 = A.to_global().cpu(), B.to_global().cpu()
    triangular_solver(side, uplo, op, Diag.NonUnit, 1.0, A, B)
    torch.cuda.synchronize()
    tri = _t(torch.tril(a) if uplo == UpLo.Lower else torch.triu(a), op)
    if side == Side.Left:
        want = torch.linalg.solve(tri, b0)
    else:
        want = torch.linalg.solve(tri.mT, b0.mT).mT
    err = (B.to_global().cpu() - want).abs().max().item()
    assert err < 1e-9 * (m + n), f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("diag", [Diag.NonUnit, Diag.Unit])
def test_trsm_lln_fastpath_gpu(dtype, diag):
    """Left-Lower-NoTrans lookahead fast path: non-divisible sizes (padded
    edge tiles), alpha scaling, unit diagonal."""
    m, n, nb, alpha = 1100, 900, 256, 0.5
    A = Matrix.create(m, m, nb, nb, dtype=dtype, device="cuda")
    B = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(A, seed=5)
    # scale so the Unit solve is well conditioned: off-diag O(1/m) keeps the
    # unit-lower inverse bounded (O(1) entries grow like c^m)
    A.storage.mul_(1.0 / m)
    mutil.set_random(B, seed=6)
    a, b0 = A.to_global().cpu(), B.to_global().cpu()
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, diag, alpha, A, B)
    torch.cuda.synchronize()
    tri = torch.tril(a)
    if diag == Diag.Unit:
        tri = tri - torch.diag_embed(tri.diagonal()) + torch.eye(m, dtype=tri.dtype)
    want = torch.linalg.solve(tri, alpha * b0)
    err = (B.to_global().cpu() - want).abs().max().item()
    assert err < 1e-9 * (m + n), f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_trmm_gpu(dtype):
    m, n, nb = 1024, 512, 256
    A = Matrix.create(m, m, nb, nb, dtype=dtype, device="cuda")
    B = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(A, seed=5)
    mutil.set_random(B, seed=6)
    a, b0 = A.to_global().cpu(), B.to_global().cpu()
    triangular_multiplication(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0, A, B)
    torch.cuda.synchronize()
    want = torch.tril(a) @ b0
    err = (B.to_global().cpu() - want).abs().max().item()
    assert err < 1e-10 * m, f"err={err}"


@pytest.mark.parametrize("side", [Side.Left, Side.Right])
def test_hemm_gpu(side):
    dtype = torch.complex128
    m, n, nb = 768, 512, 256
    k = m if side == Side.Left else n
    A = Matrix.create(k, k, nb, nb, dtype=dtype, device="cuda")
    B = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
    C = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian(A, seed=7)
    mutil.set_random(B, seed=8)
    mutil.set_random(C, seed=9)
    a, b, c0 = A.to_global().cpu(), B.to_global().cpu(), C.to_global().cpu()
    hermitian_multiplication(side, UpLo.Lower, 1.0, A, B, 0.5, C)
    torch.cuda.synchronize()
    h = _herm(a)
    want = (h @ b if side == Side.Left else b @ h) + 0.5 * c0
    err = (C.to_global().cpu() - want).abs().max().item()
    assert err < 1e-10 * (m + n), f"err={err}"


@pytest.mark.parametrize("opA,opB", [(Op.NoTrans, Op.NoTrans), (Op.ConjTrans, Op.NoTrans)])
def test_gemm_gpu(opA, opB):
    dtype = torch.complex128
    m, n, kk, nb = 512, 768, 1024, 256
    sa = (m, kk) if opA is Op.NoTrans else (kk, m)
    A = Matrix.create(*sa, nb, nb, dtype=dtype, device="cuda")
    B = Matrix.create(kk, n, nb, nb, dtype=dtype, device="cuda")
    C = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random(A, seed=1)
    mutil.set_random(B, seed=2)
    mutil.set_random(C, seed=3)
    want = 2.0 * _t(A.to_global().cpu(), opA) @ B.to_global().cpu() + C.to_global().cpu()
    general_multiplication(opA, opB, 2.0, A, B, 1.0, C)
    torch.cuda.synchronize()
    err = (C.to_global().cpu() - want).abs().max().item()
    assert err < 1e-9 * kk, f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_trtri_potri_gpu(dtype):
    n, nb = 1024, 256
    mat = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=11)
    a = mat.to_global().cpu()
    L = torch.linalg.cholesky(a)
    mat.set_from_global(L.cuda())
    inverse_from_cholesky_factor(UpLo.Lower, mat)
    torch.cuda.synchronize()
    want = torch.linalg.inv(a)
    err = (torch.tril(mat.to_global().cpu()) - torch.tril(want)).abs().max().item()
    assert err < 1e-8 * n, f"err={err}"


def test_trtri_gpu():
    n, nb = 768, 256
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=12)
    a = mat.to_global().cpu()
    triangular_inverse(UpLo.Lower, Diag.NonUnit, mat)
    torch.cuda.synchronize()
    want = torch.linalg.inv(torch.tril(a))
    err = (torch.tril(mat.to_global().cpu()) - torch.tril(want)).abs().max().item()
    assert err < 1e-9 * n, f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_hegst_gpu(dtype):
    n, nb = 1024, 256
    A = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    L = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian(A, seed=21)
    mutil.set_random_hermitian_positive_definite(L, seed=22)
    b = L.to_global().cpu()
    Lf = torch.linalg.cholesky(b)
    L.set_from_global(Lf.cuda())
    a0 = A.to_global().cpu()
    generalized_to_standard(UpLo.Lower, A, L)
    torch.cuda.synchronize()
    linv = torch.linalg.inv(Lf)
    want = linv @ _herm(a0) @ linv.mH
    err = (_herm(A.to_global().cpu()) - want).abs().max().item()
    assert err < 1e-8 * n, f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_eigensolver_gpu(dtype):
    n, nb = 1024, 256
    mat = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian(mat, seed=31)
    a0 = _herm(mat.to_global().cpu())
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat)
    torch.cuda.synchronize()
    E = evecs.to_global().cpu()
    w = w.cpu()
    scale = max(1.0, w.abs().max().item())
    res = (a0 @ E - E @ torch.diag(w.to(E.dtype))).abs().max().item()
    assert res < 1e-9 * n * scale, f"res={res}"
    orth = (E.mH @ E - torch.eye(n, dtype=E.dtype)).abs().max().item()
    assert orth < 1e-10 * n, f"orth={orth}"
    import numpy as np
    wref = np.linalg.eigvalsh(a0.numpy())
    assert np.abs(np.sort(w.numpy()) - wref).max() < 1e-10 * n * scale


def test_gen_eigensolver_gpu():
    n, nb = 768, 256
    dtype = torch.float64
    A = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    B = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian(A, seed=41)
    mutil.set_random_hermitian_positive_definite(B, seed=42)
    a0, b0 = _herm(A.to_global().cpu()), _herm(B.to_global().cpu())
    w, evecs = hermitian_generalized_eigensolver(UpLo.Lower, A, B)
    torch.cuda.synchronize()
    E = evecs.to_global().cpu()
    w = w.cpu()
    scale = max(1.0, w.abs().max().item())
    res = (a0 @ E - b0 @ E @ torch.diag(w)).abs().max().item()
    assert res < 1e-8 * n * scale, f"res={res}"


def test_max_norm_gpu():
    mat = Matrix.create(512, 512, 128, 128, dtype=torch.float64, device="cuda")
    mutil.set_random(mat, seed=4)
    a = mat.to_global().cpu()
    assert abs(max_norm(mat) - a.abs().max().item()) < 1e-14


def test_panel_qr_kernel_gpu():
    """Cooperative whole-panel QR vs the CPU torch column loop."""
    from dlaf_amd.algs.red2band import panel_qr_
    for dtype in [torch.float64, torch.complex128, torch.float32]:
        for (m, nb) in [(1024, 128), (777, 64), (128, 128), (100, 128)]:
            g = torch.Generator().manual_seed(17)
            P0 = torch.randn(m, nb, generator=g, dtype=torch.float64)
            if dtype.is_complex:
                P0 = (P0 + 1j * torch.randn(m, nb, generator=g, dtype=torch.float64))
            P0 = P0.to(dtype)
            ncols = min(m, nb)
            Pc = P0.clone()
            tc = torch.zeros(ncols, dtype=dtype)
            panel_qr_(Pc, tc)
            Pg = P0.cuda()
            tg = torch.zeros(ncols, dtype=dtype, device="cuda")
            panel_qr_(Pg, tg)
            torch.cuda.synchronize()
            tol = 1e-4 if dtype == torch.float32 else 1e-11
            perr = (Pg.cpu() - Pc).abs().max().item()
            terr = (tg.cpu() - tc).abs().max().item()
            assert perr < tol * m, f"{dtype} {m}x{nb} P err={perr}"
            assert terr < tol * m, f"{dtype} {m}x{nb} tau err={terr}"


def test_secular_kernel_gpu():
    """HIP one-thread-per-root secular solver vs the CPU torch reference."""
    import dlaf_amd.algs.tridiag_dc as dc
    g = torch.Generator().manual_seed(23)
    for k in [5, 100, 700]:
        d = torch.sort(torch.randn(k, generator=g, dtype=torch.float64))[0]
        # well-separated + some clusters
        z = torch.randn(k, generator=g, dtype=torch.float64)
        z = z / z.norm()
        rho = 1.7
        sc, mc = dc._secular_roots(d, z, rho)             # CPU torch path
        sg, mg = dc._secular_roots(d.cuda(), z.cuda(), rho)  # HIP kernel
        torch.cuda.synchronize()
        lam_c = d[sc] + mc
        lam_g = (d.cuda()[sg] + mg).cpu()
        err = (lam_c - lam_g).abs().max().item()
        assert err < 1e-12 * max(1.0, d.abs().max().item()), f"k={k} err={err}"
        # residual of the secular equation at the GPU roots
        dd = d.unsqueeze(1) - d[sg.cpu()].unsqueeze(0)
        f = 1.0 + rho * ((z * z).unsqueeze(1) / (dd - mg.cpu().unsqueeze(0))).sum(0)
        fp = rho * ((z * z).unsqueeze(1) / (dd - mg.cpu().unsqueeze(0)) ** 2).sum(0)
        assert (f.abs() / fp).max().item() < 1e-12, f"k={k}"


def test_eigensolver_gpu_float32():
    n, nb = 768, 256
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float32, device="cuda")
    mutil.set_random_hermitian(mat, seed=35)
    a0 = _herm(mat.to_global().cpu()).to(torch.float64)
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat)
    torch.cuda.synchronize()
    E = evecs.to_global().cpu().to(torch.float64)
    w = w.cpu().to(torch.float64)
    scale = max(1.0, w.abs().max().item())
    res = (a0 @ E - E @ torch.diag(w)).abs().max().item()
    assert res < 1e-3 * n * scale, f"res={res}"
    orth = (E.mT @ E - torch.eye(n, dtype=torch.float64)).abs().max().item()
    assert orth < 1e-3 * n, f"orth={orth}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_gpu_band_chase_matches_cpu(dtype):
    """GPU wavefront chase vs the CPU chase on the same band: d/e and
    reflector store agree to rounding."""
    import os
    from dlaf_amd.algs.band2tridiag import chase_band
    os.environ["DLAF_GPU_CHASE"] = "1"
    torch.manual_seed(11)
    n, b = 1500, 64
    ld = 2 * b
    store = torch.zeros(n, ld, dtype=dtype)
    store[:, 0] = (torch.rand(n, dtype=torch.float64) + 2.0 * n).to(
        torch.float64 if not dtype.is_complex else torch.float64)
    for d in range(1, b + 1):
        col = torch.randn(n - d, dtype=torch.float64)
        if dtype.is_complex:
            col = col + 1j * torch.randn(n - d, dtype=torch.float64)
        store[: n - d, d] = col.to(dtype)
    cpu = chase_band(store.clone(), b)
    try:
        gpu = chase_band(store.clone().cuda(), b)
    finally:
        os.environ.pop("DLAF_GPU_CHASE", None)
    assert gpu.vstore.is_cuda, "GPU chase did not run (fell back to CPU)"
    de = (gpu.d.cpu() - cpu.d).abs().max().item()
    ee = (gpu.e.cpu() - cpu.e).abs().max().item()
    ve = (gpu.vstore.cpu() - cpu.vstore).abs().max().item()
    assert de < 1e-8 * n and ee < 1e-8 * n, f"d={de} e={ee}"
    # reflector rounding compounds through ~n dependent sweeps
    assert ve < 1e-6, f"vstore diff {ve}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_cholesky_upper_native_gpu(dtype):
    """Native Upper POTRF on device (lookahead schedule, no transposes)."""
    from dlaf_amd import Matrix, UpLo, cholesky_factorization
    from dlaf_amd.matrix import util as mutil
    n, nb = 1536, 512
    mat = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=13)
    A = mat.to_global()
    A = torch.tril(A) + torch.tril(A, -1).mH
    mat.set_from_global(A.clone())
    cholesky_factorization(UpLo.Upper, mat)
    U = torch.triu(mat.to_global())
    err = (U.mH @ U - A).abs().max().item()
    scale = A.abs().max().item()
    assert err < 1e-10 * n * scale, err


def test_hegst_upper_native_gpu():
    """Native Upper HEGST on device."""
    from dlaf_amd import Matrix, UpLo, generalized_to_standard
    from dlaf_amd.matrix import util as mutil
    n, nb = 1024, 256
    dtype = torch.complex128
    a = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian(a, seed=5)
    A = a.to_global()
    A = torch.tril(A) + torch.tril(A, -1).mH
    a.set_from_global(A.clone())
    u = Matrix.create(n, n, nb, nb, dtype=dtype, device="cuda")
    mutil.set_random_hermitian_positive_definite(u, seed=6)
    B = u.to_global()
    B = torch.tril(B) + torch.tril(B, -1).mH
    U = torch.linalg.cholesky(B, upper=True)
    u.set_from_global(U.clone())
    generalized_to_standard(UpLo.Upper, a, u)
    got = a.to_global()
    got = torch.triu(got) + torch.triu(got, 1).mH
    Ui = torch.linalg.solve_triangular(
        U, torch.eye(n, dtype=dtype, device="cuda"), upper=True)
    want = Ui.mH @ A @ Ui
    err = (got - want).abs().max().item()
    assert err < 1e-10 * n, err


def test_dc_rot_batch_gpu(monkeypatch):
    """Gated DLAF_DC_ROT_BATCH=1 path on device at a modest size (clean at
    n=8192 in validation; faults at n=20000 — docs/DESIGN.md): result must
    bitwise-match the default sequential rotation apply."""
    from dlaf_amd.algs.tridiag_dc import tridiagonal_eigensolver
    g = torch.Generator().manual_seed(3)
    n = 2048
    d = torch.ones(n, dtype=torch.float64)
    d[::2] = 2.0
    e = 1e-3 * torch.randn(n - 1, generator=g, dtype=torch.float64).abs()
    monkeypatch.delenv("DLAF_DC_ROT_BATCH", raising=False)
    w1, E1 = tridiagonal_eigensolver(d.clone(), e.clone(), device="cuda")
    monkeypatch.setenv("DLAF_DC_ROT_BATCH", "1")
    w2, E2 = tridiagonal_eigensolver(d.clone(), e.clone(), device="cuda")
    assert torch.equal(w1, w2)
    assert torch.equal(E1, E2)
