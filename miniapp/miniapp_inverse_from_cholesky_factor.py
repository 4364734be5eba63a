#!/usr/bin/env python3
"""POTRI miniapp (reference ``miniapp/miniapp_inverse_from_cholesky_factor.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_spd
from dlaf_amd import UpLo, cholesky_factorization, inverse_from_cholesky_factor
from dlaf_amd.types import total_ops


def setup(ctx):
    a = random_spd(ctx)
    cholesky_factorization(UpLo.Lower, a, ctx.comm_grid)
    return {"a": a}


def run(ctx, st):
    inverse_from_cholesky_factor(UpLo.Lower, st["a"], ctx.comm_grid)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, n**3 / 3, n**3 / 3)


if __name__ == "__main__":
    run_miniapp("miniapp_inverse_from_cholesky_factor", setup, run, flops)
