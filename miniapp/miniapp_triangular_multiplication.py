#!/usr/bin/env python3
"""TRMM miniapp (reference ``miniapp/miniapp_triangular_multiplication.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_spd, random_general
from dlaf_amd import Side, UpLo, Op, Diag, triangular_multiplication
from dlaf_amd.types import total_ops


def extra(p):
    p.add_argument("--n", type=int, default=0)


def setup(ctx):
    m = ctx.opts.matrix_size
    n = ctx.opts.n or m
    st = {"a": random_spd(ctx), "b": random_general(ctx, m, n)}
    if ctx.opts.check_result != "none":
        st["b0"] = st["b"].storage.clone()
    return st


def run(ctx, st):
    triangular_multiplication(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0,
                              st["a"], st["b"], ctx.comm_grid)


def flops(ctx):
    m = float(ctx.opts.matrix_size)
    n = float(ctx.opts.n or ctx.opts.matrix_size)
    add_mul = m * m * n / 2
    return total_ops(ctx.dtype, add_mul, add_mul)


def check(ctx, st, _):
    """Invert the multiply with the solver and compare to B0 (GPU-side)."""
    import torch
    from dlaf_amd import Matrix, triangular_solver
    b = st["b"]
    bx = Matrix.create(b.dist.m, b.dist.n, b.dist.mb, b.dist.nb,
                       dtype=b.dtype, device=b.device, grid=ctx.grid)
    bx.storage.copy_(b.storage)
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0,
                      st["a"], bx, ctx.comm_grid)
    diff = (bx.storage - st["b0"]).abs().max()
    scale = st["b0"].abs().max()
    if ctx.grid.distributed:
        import torch.distributed as dist
        dist.all_reduce(diff, op=dist.ReduceOp.MAX)
        dist.all_reduce(scale, op=dist.ReduceOp.MAX)
    return (diff / (ctx.opts.matrix_size * scale)).item()


if __name__ == "__main__":
    run_miniapp("miniapp_triangular_multiplication", setup, run, flops, check,
                extra=extra)
