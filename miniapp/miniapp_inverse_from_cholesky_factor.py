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
    st = {}
    if ctx.opts.check_result != "none":
        st["a0"] = a.clone()
    cholesky_factorization(UpLo.Lower, a, ctx.comm_grid)
    st["a"] = a
    return st


def run(ctx, st):
    inverse_from_cholesky_factor(UpLo.Lower, st["a"], ctx.comm_grid)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, n**3 / 3, n**3 / 3)


def check(ctx, st, _):
    """max |A0 X - I| (X = the computed inverse, lower stored)."""
    import torch
    x = st["a"].to_global()
    x = torch.tril(x) + torch.tril(x, -1).mH
    a0 = st["a0"].to_global()
    a0 = torch.tril(a0) + torch.tril(a0, -1).mH
    n = a0.shape[0]
    eye = torch.eye(n, dtype=a0.dtype, device=a0.device)
    return (a0 @ x - eye).abs().max().item()


if __name__ == "__main__":
    run_miniapp("miniapp_inverse_from_cholesky_factor", setup, run, flops, check)
