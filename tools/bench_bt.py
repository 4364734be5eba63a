#!/usr/bin/env python3
"""bt_band_to_tridiag A/B: group kernel vs torch GEMM chain."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from dlaf_amd.algs import band2tridiag as b2t
from bench_chase_gpu import make_band

n = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
b = int(sys.argv[2]) if len(sys.argv) > 2 else 64
tc = sys.argv[3] if len(sys.argv) > 3 else "d"
modes = sys.argv[4] if len(sys.argv) > 4 else "01"
dtype = {"d": torch.float64, "z": torch.complex128}[tc]
os.environ["DLAF_GPU_CHASE"] = "1" if tc == "z" else "0"
band = make_band(n, b, dtype, "cuda")
tri = b2t.chase_band(band, b)
E0 = torch.randn(n, n, dtype=torch.float64, device="cuda").to(dtype)
for mode in modes:
    os.environ["DLAF_BT_KERNEL"] = mode
    E = E0.clone()
    b2t.bt_band_to_tridiagonal(E, tri)   # warm
    torch.cuda.synchronize()
    E = E0.clone()
    t0 = time.perf_counter()
    b2t.bt_band_to_tridiagonal(E, tri)
    torch.cuda.synchronize()
    print(f"bt {tc} n={n} b={b} kernel={mode}: {time.perf_counter()-t0:.2f} s", flush=True)
