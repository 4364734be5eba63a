"""Local CPU benchmark of the band chase (correctness + wall time)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dlaf_amd.ops._ext import get_ext
from dlaf_amd.algs.band2tridiag import _slot_counts

def run(n=8192, b=64, nthreads=0, seed=3, check=False):
    torch.manual_seed(seed)
    ld = 2 * b
    band = torch.zeros(n, ld, dtype=torch.float64)
    band[:, 0] = torch.rand(n) + 2.0 * n
    for d in range(1, b + 1):
        band[: n - d, d] = torch.randn(n - d)
    counts = _slot_counts(n, b)
    offs = torch.zeros(n, dtype=torch.int64)
    offs[1:] = torch.cumsum(counts, 0)[:-1]
    vstore = torch.zeros(int(counts.sum()), b + 1, dtype=torch.float64)
    ext = get_ext()
    ref = None
    if check:
        b1, v1 = band.clone(), vstore.clone()
        ext.band_chase(b1, b, v1, offs, 1)
        ref = (b1, v1)
    t0 = time.perf_counter()
    ext.band_chase(band, b, vstore, offs, nthreads)
    dt = time.perf_counter() - t0
    tag = f"n={n} b={b} nt={nthreads}"
    print(f"chase {tag}: {dt:.3f}s", flush=True)
    if ref is not None:
        db = (band - ref[0]).abs().max().item()
        dv = (vstore - ref[1]).abs().max().item()
        print(f"  vs sequential: band {db:.1e} vstore {dv:.1e}", flush=True)

if __name__ == "__main__":
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=8192)
    p.add_argument("--b", type=int, default=64)
    p.add_argument("--nt", type=int, default=0)
    p.add_argument("--check", action="store_true")
    a = p.parse_args()
    run(a.n, a.b, a.nt, check=a.check)
