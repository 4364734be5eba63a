#!/usr/bin/env python3
"""reduction_to_band miniapp (reference ``miniapp/miniapp_reduction_to_band.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_herm
from dlaf_amd.algs.red2band import reduction_to_band
from dlaf_amd.algs.eigensolver import get_band_size
from dlaf_amd.types import total_ops


def extra(p):
    p.add_argument("--band-size", type=int, default=0)


def setup(ctx):
    return {"a": random_herm(ctx)}


def run(ctx, st):
    band = ctx.opts.band_size or get_band_size(ctx.opts.block_size)
    reduction_to_band(st["a"], band)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, 2 * n**3 / 3, 2 * n**3 / 3)


if __name__ == "__main__":
    run_miniapp("miniapp_reduction_to_band", setup, run, flops, extra=extra)
