#!/usr/bin/env python3
"""Kernel microbenchmarks on one MI355X: fused GEMM TF/s, potrf/trtri latency."""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from dlaf_amd.ops import tile_ops as ops
from dlaf_amd.types import Op


def bench_gemm(dtype=torch.float64, nb=512, ntiles=128, iters=10, opB=Op.Trans, inplace=False):
    dev = "cuda"
    A = torch.randn(ntiles, nb, nb, dtype=dtype, device=dev) if not dtype.is_complex else (
        torch.randn(ntiles, nb, nb, dtype=torch.float64, device=dev)
        + 1j * torch.randn(ntiles, nb, nb, dtype=torch.float64, device=dev)).to(dtype)
    C = torch.zeros_like(A)
    ts = nb * nb
    offs = [i * ts for i in range(ntiles)]
    descs = ops.make_descs(offs, offs, offs)
    dt = torch.from_numpy(descs).to(dev)
    # warmup
    for _ in range(3):
        ops.gemm_fused(C, A, A, dt, nb, nb, nb, nb, nb, nb, Op.NoTrans, opB, -1.0, 1.0, inplace=inplace)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.gemm_fused(C, A, A, dt, nb, nb, nb, nb, nb, nb, Op.NoTrans, opB, -1.0, 1.0, inplace=inplace)
    torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / iters
    flops = 2.0 * ntiles * nb * nb * nb * (4 if dtype.is_complex else 1)
    print(f"gemm {dtype} nb={nb} ntiles={ntiles} opB={opB} bn128={inplace}: {dt_s*1e3:.2f} ms  "
          f"{flops/dt_s/1e12:.2f} TFLOP/s")


def bench_torch_bmm(dtype=torch.float64, nb=512, ntiles=128, iters=10):
    """rocBLAS batched GEMM baseline for the same shape (library comparison)."""
    dev = "cuda"
    A = torch.randn(ntiles, nb, nb, dtype=dtype, device=dev) if not dtype.is_complex else (
        torch.randn(ntiles, nb, nb, dtype=torch.float64, device=dev)
        + 1j * torch.randn(ntiles, nb, nb, dtype=torch.float64, device=dev)).to(dtype)
    C = torch.zeros_like(A)
    for _ in range(3):
        torch.bmm(A, A.mT, out=C)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        torch.bmm(A, A.mT, out=C)
    torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / iters
    flops = 2.0 * ntiles * nb * nb * nb * (4 if dtype.is_complex else 1)
    print(f"torch.bmm {dtype} nb={nb} ntiles={ntiles}: {dt_s*1e3:.2f} ms  "
          f"{flops/dt_s/1e12:.2f} TFLOP/s")


def bench_big_dgemm(dtype=torch.float64, n=16384, iters=5):
    """Single large library DGEMM: the machine speed-of-light reference."""
    a = torch.randn(n, n, dtype=dtype, device="cuda")
    c = torch.zeros_like(a)
    for _ in range(2):
        torch.mm(a, a.mT, out=c)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        torch.mm(a, a.mT, out=c)
    torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / iters
    print(f"torch.mm {dtype} n={n}: {dt_s*1e3:.1f} ms  {2*n**3/dt_s/1e12:.2f} TFLOP/s")


def bench_potrf(dtype=torch.float64, nb=512, iters=20):
    a = torch.randn(nb, nb, dtype=dtype, device="cuda")
    a = a @ a.mH + nb * torch.eye(nb, dtype=dtype, device="cuda")
    t = a.clone()
    dinv = ops.dinv_workspace(nb, dtype, "cuda")
    for _ in range(3):
        t.copy_(a)
        ops.potrf_tile(t, dinv)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        t.copy_(a)
        ops.potrf_tile(t, dinv)
    torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / iters
    print(f"potrf_tile {dtype} nb={nb}: {dt_s*1e6:.0f} us")


def bench_trtri(dtype=torch.float64, nb=512, iters=20):
    L = torch.tril(torch.randn(nb, nb, dtype=dtype, device="cuda")) + \
        2 * nb * torch.eye(nb, dtype=dtype, device="cuda")
    T = torch.empty_like(L)
    for _ in range(3):
        ops.tri_inverse_full(L, lower=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.tri_inverse_full(L, lower=True)
    torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / iters
    print(f"trtri_tile {dtype} nb={nb}: {dt_s*1e6:.0f} us")



def bench_lib_batched(dtype=torch.float64, nb=512, nt=63, iters=5):
    """rocBLAS batched vs fused kernel on the POTRF trailing geometry:
    C[i,j] -= P[i] @ P[j]^T over the lower triangle (i >= j)."""
    from dlaf_amd.ops import tile_ops as ops
    from dlaf_amd.types import Op
    dev = "cuda"
    panel = torch.randn(nt, nb, nb, dtype=dtype, device=dev) if not dtype.is_complex \
        else torch.randn(nt, nb, nb, dtype=dtype, device=dev)
    items = [(i, j) for j in range(nt) for i in range(j, nt)]
    ntc = len(items)
    C = torch.randn(ntc, nb, nb, dtype=dtype, device=dev)
    C2 = C.clone()
    offc = [e * nb * nb for e in range(ntc)]
    offa = [i * nb * nb for i, _ in items]
    offb = [j * nb * nb for _, j in items]
    pC, pA, pB = (ops.ptr_array(C, offc), ops.ptr_array(panel, offa),
                  ops.ptr_array(panel, offb))
    pC2 = ops.ptr_array(C2, offc)
    d = ops.make_descs(offc, offa, offb)
    opB = Op.ConjTrans if dtype.is_complex else Op.Trans
    # numerics: lib vs fused
    ops.gemm_fused(C.reshape(-1), panel.reshape(-1), panel.reshape(-1), d,
                   nb, nb, nb, nb, nb, nb, Op.NoTrans, opB, -1.0, 1.0)
    ops.gemm_batched_lib(C2, pC2, pA, pB, nb, nb, nb, nb, nb, nb,
                         Op.NoTrans, opB, -1.0, 1.0)
    torch.cuda.synchronize()
    print(f"lib vs fused max diff: {(C - C2).abs().max().item():.3e}")
    import time
    for tag, fn in (
        ("fused", lambda: ops.gemm_fused(C.reshape(-1), panel.reshape(-1),
                                         panel.reshape(-1), d, nb, nb, nb, nb, nb, nb,
                                         Op.NoTrans, opB, -1.0, 1.0)),
        ("rocblas_batched", lambda: ops.gemm_batched_lib(
            C, pC, pA, pB, nb, nb, nb, nb, nb, nb, Op.NoTrans, opB, -1.0, 1.0)),
    ):
        fn(); torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        dt_s = (time.perf_counter() - t0) / iters
        mul = 4 if dtype.is_complex else 1
        fl = 2.0 * nb**3 * ntc * mul
        print(f"trailing {tag} {dtype} nb={nb} tiles={ntc}: {dt_s*1e3:.2f} ms  "
              f"{fl/dt_s/1e12:.2f} TFLOP/s", flush=True)

if __name__ == "__main__":
    print(torch.cuda.get_device_name(0))
    bench_gemm(torch.float64, 512, 128)
    bench_gemm(torch.float64, 512, 1024, iters=5)
    bench_gemm(torch.float32, 512, 128)
    bench_gemm(torch.complex128, 512, 64)
    bench_torch_bmm(torch.float64, 512, 128)
    bench_torch_bmm(torch.float64, 512, 1024, iters=5)
    if "--lib" in sys.argv:
        bench_lib_batched(torch.float64)
        bench_lib_batched(torch.complex128, nt=44)
    bench_big_dgemm(torch.float64, 16384)
    bench_potrf(torch.float64)
    bench_potrf(torch.complex128, 512)
    bench_trtri(torch.float64)
