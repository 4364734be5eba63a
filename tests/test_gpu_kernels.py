"""GPU kernel numerics tests: hand-written CDNA4 kernels vs torch CPU reference.

Every HIP kernel is compared against a plain PyTorch reference of the same op
computed in fp64 (or the op's own precision when that IS fp64).
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from dlaf_amd.types import Op  # noqa: E402
from dlaf_amd.ops import tile_ops as ops  # noqa: E402

DTYPES = [torch.float64, torch.float32, torch.complex128, torch.complex64]


def _tol(dtype, k=512):
    if dtype in (torch.float32, torch.complex64):
        return 1e-4 * max(1, k // 64)
    return 1e-12 * max(1, k // 64)


def _rand(shape, dtype, device="cuda"):
    if dtype.is_complex:
        rd = torch.float64 if dtype == torch.complex128 else torch.float32
        return torch.complex(
            torch.randn(shape, dtype=rd), torch.randn(shape, dtype=rd)
        ).to(device=device, dtype=dtype)
    return torch.randn(shape, dtype=dtype, device=device)


def _ref_op(x, op):
    if op is Op.NoTrans:
        return x
    if op is Op.Trans:
        return x.mT
    return x.mH


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("opA", [Op.NoTrans, Op.Trans, Op.ConjTrans])
@pytest.mark.parametrize("opB", [Op.NoTrans, Op.Trans, Op.ConjTrans])
def test_batch_gemm_ops(dtype, opA, opB):
    torch.manual_seed(0)
    M, N, K = 256, 128, 192
    # physical shapes depend on op
    a_shape = (M, K) if opA is Op.NoTrans else (K, M)
    b_shape = (K, N) if opB is Op.NoTrans else (N, K)
    A = _rand(a_shape, dtype)
    B = _rand(b_shape, dtype)
    C = _rand((M, N), dtype)
    C0 = C.clone()
    alpha, beta = (1.5 - 0.5j, 0.25 + 1j) if dtype.is_complex else (1.5, 0.25)
    descs = ops.make_descs([0], [0], [0])
    ops.gemm_fused(C, A, B, descs, M, N, K, a_shape[1], b_shape[1], N, opA, opB, alpha, beta)
    torch.cuda.synchronize()
    ref = alpha * (_ref_op(A.cpu().to(torch.promote_types(dtype, torch.complex128 if dtype.is_complex else torch.float64)), opA)
                   @ _ref_op(B.cpu().to(torch.promote_types(dtype, torch.complex128 if dtype.is_complex else torch.float64)), opB)) \
        + beta * C0.cpu().to(torch.promote_types(dtype, torch.complex128 if dtype.is_complex else torch.float64))
    err = (C.cpu().to(ref.dtype) - ref).abs().max().item()
    scale = ref.abs().max().item() + 1
    assert err <= _tol(dtype, K) * scale, f"err={err} scale={scale}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("M,N,K", [(64, 64, 64), (100, 96, 120), (33, 17, 5), (512, 512, 512), (128, 64, 1)])
def test_batch_gemm_shapes(dtype, M, N, K):
    torch.manual_seed(1)
    A = _rand((M, K), dtype)
    B = _rand((K, N), dtype)
    C = torch.zeros((M, N), dtype=dtype, device="cuda")
    descs = ops.make_descs([0], [0], [0])
    ops.gemm_fused(C, A, B, descs, M, N, K, K, N, N, Op.NoTrans, Op.NoTrans, 1.0, 0.0)
    torch.cuda.synchronize()
    ref = A.cpu().to(torch.complex128 if dtype.is_complex else torch.float64) @ \
        B.cpu().to(torch.complex128 if dtype.is_complex else torch.float64)
    err = (C.cpu().to(ref.dtype) - ref).abs().max().item()
    assert err <= _tol(dtype, K) * (ref.abs().max().item() + 1), f"err={err}"


def test_batch_gemm_multi_desc_and_ktiles():
    torch.manual_seed(2)
    nb = 64
    nt = 3
    # A: row of k-tiles, B: column of k-tiles, C: single tile accumulated over 2 k-tiles
    A = torch.randn(nt, nb, nb, dtype=torch.float64, device="cuda")
    B = torch.randn(nt, nb, nb, dtype=torch.float64, device="cuda")
    C = torch.zeros(2, nb, nb, dtype=torch.float64, device="cuda")
    ts = nb * nb
    descs = ops.make_descs([0, ts], [0, ts], [0, ts], ktiles=2, a_kstride=ts, b_kstride=ts)
    ops.gemm_fused(C, A, B, descs, nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.NoTrans, 1.0, 0.0)
    torch.cuda.synchronize()
    for d in range(2):
        ref = A[d].cpu() @ B[d].cpu() + A[d + 1].cpu() @ B[d + 1].cpu()
        err = (C[d].cpu() - ref).abs().max().item()
        assert err < 1e-10, f"desc {d}: err={err}"


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("n", [64, 128, 100, 512, 500])
def test_potrf_tile(dtype, n):
    if dtype.is_complex and n == 100:
        n = 60  # complex block size is 64
    torch.manual_seed(3)
    a = _rand((n, n), dtype).cpu()
    a = a @ a.mH + n * torch.eye(n, dtype=dtype)
    tile = a.to("cuda")
    dinv = ops.potrf_tile(tile)
    torch.cuda.synchronize()
    ref = torch.linalg.cholesky(a.to(torch.complex128 if dtype.is_complex else torch.float64))
    got = torch.tril(tile.cpu().to(ref.dtype))
    err = (got - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err <= _tol(dtype, n) * scale * 10, f"err={err}"
    # dinv blocks invert the diagonal blocks of the factor
    bsz = dinv.shape[-1]
    d0 = dinv[0].cpu().to(ref.dtype)
    bs = min(bsz, n)
    prod = d0[:bs, :bs] @ ref[:bs, :bs]
    err = (prod - torch.eye(bs, dtype=ref.dtype)).abs().max().item()
    assert err <= _tol(dtype, n) * 100, f"dinv err={err}"


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("n", [64, 128, 512, 300])
def test_trtri_lower(dtype, n):
    torch.manual_seed(4)
    L = torch.tril(_rand((n, n), dtype)) + 2 * n ** 0.5 * torch.eye(n, dtype=dtype, device="cuda")
    T = torch.empty_like(L)
    ops.trtri_tile(L, T)
    torch.cuda.synchronize()
    rdt = torch.complex128 if dtype.is_complex else torch.float64
    prod = T.cpu().to(rdt) @ L.cpu().to(rdt)
    err = (prod - torch.eye(n, dtype=rdt)).abs().max().item()
    assert err <= _tol(dtype, n) * 100, f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128, torch.float32])
def test_trsm_panel(dtype):
    torch.manual_seed(5)
    nb = 256 if not dtype.is_complex else 128
    ntiles = 3
    L = torch.tril(_rand((nb, nb), dtype)) + 2 * nb * torch.eye(nb, dtype=dtype, device="cuda")
    panel = _rand((ntiles, nb, nb), dtype)
    ref_in = panel.clone().cpu()
    # dinv of L
    dinv = ops.dinv_workspace(nb, dtype, "cuda")
    bsz = dinv.shape[-1]
    ext = ops.get_ext()
    for d in range((nb + bsz - 1) // bsz):
        c0 = d * bsz
        bs = min(bsz, nb - c0)
        ext.trtri_lower(L[c0:, c0:], dinv[d], bs, L.stride(0), bsz, False)
    offs = [i * nb * nb for i in range(ntiles)]
    ops.trsm_panel_right_lowerH(panel, offs, L, dinv, nb, nb, nb)
    torch.cuda.synchronize()
    rdt = torch.complex128 if dtype.is_complex else torch.float64
    Lh = L.cpu().to(rdt).mH
    for i in range(ntiles):
        ref = torch.linalg.solve_triangular(Lh, ref_in[i].to(rdt), upper=True, left=False)
        err = (panel[i].cpu().to(rdt) - ref).abs().max().item()
        scale = ref.abs().max().item() + 1
        assert err <= _tol(dtype, nb) * scale * 100, f"tile {i} err={err}"


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("opA", [Op.NoTrans, Op.Trans, Op.ConjTrans])
@pytest.mark.parametrize("opB", [Op.NoTrans, Op.Trans, Op.ConjTrans])
def test_gemm_v2_fulltile_ktiles(dtype, opA, opB):
    """v2 glds fast path: full-tile multi-desc batch with K-tile accumulation
    (the hot trailing-update shape) vs fp64 torch reference."""
    torch.manual_seed(3)
    nb = 128 if not dtype.is_complex else 64
    M = N = nb
    K = nb
    nt = 4  # operand tiles
    ts = nb * nb
    A = _rand((nt, nb, nb), dtype)
    B = _rand((nt, nb, nb), dtype)
    C = _rand((2, nb, nb), dtype)
    C0 = C.clone()
    # two descs: desc0 accumulates ktiles 0..1, desc1 ktiles 2..3
    descs = ops.make_descs([0, ts], [0, 2 * ts], [0, 2 * ts], ktiles=2,
                           a_kstride=ts, b_kstride=ts)
    alpha, beta = (1.5 - 0.5j, 0.25 + 1j) if dtype.is_complex else (-1.0, 1.0)
    ops.gemm_fused(C, A, B, descs, M, N, K, nb, nb, nb, opA, opB, alpha, beta)
    torch.cuda.synchronize()
    hp = torch.complex128 if dtype.is_complex else torch.float64
    for d in range(2):
        ref = beta * C0[d].cpu().to(hp)
        for kt in range(2):
            t = 2 * d + kt
            ref = ref + alpha * (_ref_op(A[t].cpu().to(hp), opA)
                                 @ _ref_op(B[t].cpu().to(hp), opB))
        err = (C[d].cpu().to(hp) - ref).abs().max().item()
        scale = ref.abs().max().item() + 1
        assert err <= _tol(dtype, 2 * K) * scale, f"desc {d}: err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.float32])
def test_gemm_v2_wide(dtype):
    """v2 at a multi-block C (512x384) with rectangular K."""
    torch.manual_seed(4)
    M, N, K = 512, 384, 256
    A = _rand((M, K), dtype)
    B = _rand((K, N), dtype)
    C = torch.zeros((M, N), dtype=dtype, device="cuda")
    descs = ops.make_descs([0], [0], [0])
    ops.gemm_fused(C, A, B, descs, M, N, K, K, N, N, Op.NoTrans, Op.NoTrans, 1.0, 0.0)
    torch.cuda.synchronize()
    hp = torch.float64
    ref = A.cpu().to(hp) @ B.cpu().to(hp)
    err = (C.cpu().to(hp) - ref).abs().max().item()
    assert err <= _tol(dtype, K) * (ref.abs().max().item() + 1), f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_gemm_inplace_wide_n(dtype):
    """In-place X = X @ op(B) with N wider than one kernel column block:
    exercises the scratch-staging path in gemm_fused (round-2 race fix)."""
    torch.manual_seed(5)
    nb = 256
    nt = 3
    X = _rand((nt, nb, nb), dtype)
    Bm = _rand((nb, nb), dtype)
    X0 = X.clone()
    ts = nb * nb
    offs = [i * ts for i in range(nt)]
    descs = ops.make_descs(offs, offs, [0] * nt)
    ops.gemm_fused(X, X, Bm, descs, nb, nb, nb, nb, nb, nb,
                   Op.NoTrans, Op.ConjTrans if dtype.is_complex else Op.Trans,
                   1.0, 0.0, inplace=True)
    torch.cuda.synchronize()
    hp = torch.complex128 if dtype.is_complex else torch.float64
    for i in range(nt):
        ref = X0[i].cpu().to(hp) @ (Bm.cpu().to(hp).mH if dtype.is_complex
                                    else Bm.cpu().to(hp).mT)
        err = (X[i].cpu().to(hp) - ref).abs().max().item()
        assert err <= _tol(dtype, nb) * (ref.abs().max().item() + 1), f"{i}: {err}"


@pytest.mark.parametrize("tc", ["d", "z"])
def test_bt_apply_group_kernel_vs_torch(tc):
    """Whole-group bt window-chain kernel (csrc/bt_apply.hip) vs the torch
    GEMM chain on identical inputs."""
    import os
    from dlaf_amd.algs import band2tridiag as b2t
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "tools"))
    from bench_chase_gpu import make_band
    dtype = torch.float64 if tc == "d" else torch.complex128
    n, b = 700, 16
    os.environ["DLAF_GPU_CHASE"] = "0"
    band = make_band(n, b, dtype, "cuda")
    tri = b2t.chase_band(band, b)
    nE = 100
    E0 = _rand((n, nE), dtype)
    outs = {}
    for mode in ("0", "1"):
        os.environ["DLAF_BT_KERNEL"] = mode
        E = E0.clone()
        b2t.bt_band_to_tridiagonal(E, tri)
        torch.cuda.synchronize()
        outs[mode] = E.cpu()
    os.environ.pop("DLAF_BT_KERNEL", None)
    err = (outs["0"] - outs["1"]).abs().max().item()
    scale = outs["0"].abs().max().item() + 1
    assert err < 1e-11 * scale * n, err


def test_bt_apply_group_tail_split():
    """The CW=32 tail launch (column chunks > 256) against the torch chain."""
    import os
    from dlaf_amd.algs import band2tridiag as b2t
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "tools"))
    from bench_chase_gpu import make_band
    n, b = 500, 16
    os.environ["DLAF_GPU_CHASE"] = "0"
    band = make_band(n, b, torch.float64, "cuda")
    tri = b2t.chase_band(band, b)
    nE = 16448  # 257 64-col chunks -> 256 CW64 + 2 CW32
    E0 = torch.randn(n, nE, dtype=torch.float64, device="cuda")
    outs = {}
    for mode in ("0", "1"):
        os.environ["DLAF_BT_KERNEL"] = mode
        E = E0.clone()
        b2t.bt_band_to_tridiagonal(E, tri)
        torch.cuda.synchronize()
        outs[mode] = E.cpu()
    os.environ.pop("DLAF_BT_KERNEL", None)
    err = (outs["0"] - outs["1"]).abs().max().item()
    scale = outs["0"].abs().max().item() + 1
    assert err < 1e-11 * scale * n, err
