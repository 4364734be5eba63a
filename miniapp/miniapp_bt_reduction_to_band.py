#!/usr/bin/env python3
"""bt_reduction_to_band miniapp (reference ``miniapp/miniapp_bt_reduction_to_band.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from _harness import run_miniapp, random_herm
from dlaf_amd.algs.red2band import reduction_to_band, bt_reduction_to_band
from dlaf_amd.algs.eigensolver import get_band_size


def extra(p):
    p.add_argument("--band-size", type=int, default=0)


def setup(ctx):
    a = random_herm(ctx)
    band = ctx.opts.band_size or get_band_size(ctx.opts.block_size)
    refl = reduction_to_band(a, band)
    n = ctx.opts.matrix_size
    E = torch.randn(n, n, dtype=torch.float64, device=ctx.device).to(ctx.dtype)
    return {"a": a, "refl": refl, "E": E}


def run(ctx, st):
    bt_reduction_to_band(st["E"], st["a"], st["refl"])


if __name__ == "__main__":
    run_miniapp("miniapp_bt_reduction_to_band", setup, run, lambda ctx: None, extra=extra)
