#!/usr/bin/env python3
"""Cholesky miniapp (reference ``miniapp/miniapp_cholesky.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from _harness import run_miniapp, random_spd
from dlaf_amd import UpLo, cholesky_factorization
from dlaf_amd.types import total_ops


def setup(ctx):
    a = random_spd(ctx)
    return {"ref": a.clone(), "a": a}


def run(ctx, st):
    cholesky_factorization(UpLo.Lower, st["a"], ctx.comm_grid)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, n**3 / 6, n**3 / 6)


def check(ctx, st, _):
    # || A - L L^H ||_max / ||A||_max  (miniapp_cholesky.cpp:205-280 style)
    a = st["ref"].to_global()
    L = torch.tril(st["a"].to_global())
    return ((a - L @ L.mH).abs().max() / a.abs().max()).item()


if __name__ == "__main__":
    run_miniapp("miniapp_cholesky", setup, run, flops, check)
