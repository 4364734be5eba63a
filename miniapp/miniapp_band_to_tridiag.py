#!/usr/bin/env python3
"""band_to_tridiag miniapp (reference ``miniapp/miniapp_band_to_tridiag.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_herm
from dlaf_amd import UpLo
from dlaf_amd.algs.band2tridiag import band_to_tridiagonal
from dlaf_amd.algs.eigensolver import get_band_size


def extra(p):
    p.add_argument("--band-size", type=int, default=0)


def setup(ctx):
    return {"a": random_herm(ctx)}


def run(ctx, st):
    band = ctx.opts.band_size or get_band_size(ctx.opts.block_size)
    band_to_tridiagonal(UpLo.Lower, band, st["a"])


if __name__ == "__main__":
    run_miniapp("miniapp_band_to_tridiag", setup, run, lambda ctx: None, extra=extra)
