#!/usr/bin/env python3
"""bt_band_to_tridiag miniapp (reference ``miniapp/miniapp_bt_band_to_tridiag.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from _harness import run_miniapp, random_herm
from dlaf_amd import UpLo
from dlaf_amd.algs.band2tridiag import band_to_tridiagonal, bt_band_to_tridiagonal
from dlaf_amd.algs.eigensolver import get_band_size


def extra(p):
    p.add_argument("--band-size", type=int, default=0)


def setup(ctx):
    a = random_herm(ctx)
    band = ctx.opts.band_size or get_band_size(ctx.opts.block_size)
    tri = band_to_tridiagonal(UpLo.Lower, band, a)
    n = ctx.opts.matrix_size
    E = torch.randn(n, n, dtype=torch.float64, device=ctx.device).to(ctx.dtype)
    return {"tri": tri, "E": E}


def run(ctx, st):
    bt_band_to_tridiagonal(st["E"], st["tri"])


if __name__ == "__main__":
    run_miniapp("miniapp_bt_band_to_tridiag", setup, run, lambda ctx: None, extra=extra)
