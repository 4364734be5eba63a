#!/usr/bin/env python3
"""A/B the GPU bulge chase against the CPU wavefront on a synthetic band.

Usage: bench_chase_gpu.py [n] [b] [dtype s|d|c|z]
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from dlaf_amd.algs import band2tridiag as b2t


def make_band(n, b, dtype, device):
    ld = 2 * b
    g = torch.Generator().manual_seed(7)
    if dtype.is_complex:
        rd = torch.float64 if dtype == torch.complex128 else torch.float32
        band = (torch.randn(n, ld, generator=g, dtype=rd)
                + 1j * torch.randn(n, ld, generator=g, dtype=rd)).to(dtype)
        band[:, 0] = band[:, 0].real.to(dtype)
    else:
        band = torch.randn(n, ld, generator=g, dtype=dtype)
    band[:, b + 1:] = 0
    for j in range(n):
        band[j, max(0, n - j):] = 0
    return band.to(device)


def run(n=20000, b=64, tc="d"):
    dtype = {"d": torch.float64, "z": torch.complex128,
             "s": torch.float32, "c": torch.complex64}[tc]
    dev = "cuda"
    res = {}
    for mode in ("0", "1"):
        os.environ["DLAF_GPU_CHASE"] = mode
        band = make_band(n, b, dtype, dev)
        t0 = time.perf_counter()
        tri = b2t.chase_band(band, b)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        res[mode] = (dt, tri.d.cpu(), tri.e.cpu())
        print(f"chase {tc} n={n} b={b} gpu={mode}: {dt:.2f} s", flush=True)
    dd = (res["0"][1] - res["1"][1]).abs().max().item()
    de = (res["0"][2].abs() - res["1"][2].abs()).abs().max().item()
    print(f"  d diff {dd:.3e}  |e| diff {de:.3e}")


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 20000
    b = int(sys.argv[2]) if len(sys.argv) > 2 else 64
    tc = sys.argv[3] if len(sys.argv) > 3 else "d"
    run(n, b, tc)
