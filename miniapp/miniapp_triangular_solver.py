#!/usr/bin/env python3
"""TRSM miniapp (reference ``miniapp/miniapp_triangular_solver.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_spd, random_general
from dlaf_amd import Side, UpLo, Op, Diag, triangular_solver
from dlaf_amd.types import total_ops


def extra(p):
    p.add_argument("--n", type=int, default=0, help="B columns (default m)")


def setup(ctx):
    m = ctx.opts.matrix_size
    n = ctx.opts.n or m
    return {"a": random_spd(ctx), "b": random_general(ctx, m, n), "n": n}


def run(ctx, st):
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0,
                      st["a"], st["b"], ctx.comm_grid)


def flops(ctx):
    m = float(ctx.opts.matrix_size)
    n = float(ctx.opts.n or ctx.opts.matrix_size)
    add_mul = m * m * n / 2
    return total_ops(ctx.dtype, add_mul, add_mul)


if __name__ == "__main__":
    run_miniapp("miniapp_triangular_solver", setup, run, flops, extra=extra)
