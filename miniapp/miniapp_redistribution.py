#!/usr/bin/env python3
"""Redistribution miniapp (reference ``miniapp/miniapp_redistribution.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_general
from dlaf_amd import Matrix
from dlaf_amd.algs.redistribute import redistribute


def extra(p):
    p.add_argument("--block-size-to", type=int, default=128)


def setup(ctx):
    n = ctx.opts.matrix_size
    src = random_general(ctx, n, n)
    dst = Matrix.create(n, n, ctx.opts.block_size_to, ctx.opts.block_size_to,
                        dtype=ctx.dtype, device=ctx.device, grid=ctx.grid)
    return {"src": src, "dst": dst}


def run(ctx, st):
    redistribute(st["src"], st["dst"])


if __name__ == "__main__":
    run_miniapp("miniapp_redistribution", setup, run, lambda ctx: None, extra=extra)
