#!/usr/bin/env python3
"""TRTRI miniapp (reference ``miniapp/miniapp_triangular_inverse.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_spd
from dlaf_amd import UpLo, Diag, triangular_inverse
from dlaf_amd.types import total_ops


def setup(ctx):
    return {"a": random_spd(ctx)}


def run(ctx, st):
    triangular_inverse(UpLo.Lower, Diag.NonUnit, st["a"], ctx.comm_grid)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, n**3 / 6, n**3 / 6)


if __name__ == "__main__":
    run_miniapp("miniapp_triangular_inverse", setup, run, flops)
