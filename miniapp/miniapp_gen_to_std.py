#!/usr/bin/env python3
"""HEGST miniapp (reference ``miniapp/miniapp_gen_to_std.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_spd, random_herm
from dlaf_amd import UpLo, generalized_to_standard, cholesky_factorization
from dlaf_amd.types import total_ops


def setup(ctx):
    a = random_herm(ctx)
    l = random_spd(ctx)
    cholesky_factorization(UpLo.Lower, l, ctx.comm_grid)
    st = {"a": a, "l": l}
    if ctx.opts.check_result != "none":
        st["a0"] = a.clone()
    return st


def run(ctx, st):
    generalized_to_standard(UpLo.Lower, st["a"], st["l"], ctx.comm_grid)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, n**3 / 2, n**3 / 2)


def check(ctx, st, _):
    """max |L B L^H - A0| / |A0|_max (B = the computed standard-form matrix)."""
    import torch
    L = torch.tril(st["l"].to_global())
    b = st["a"].to_global()
    b = torch.tril(b) + torch.tril(b, -1).mH
    a0 = st["a0"].to_global()
    a0 = torch.tril(a0) + torch.tril(a0, -1).mH
    return ((L @ b @ L.mH - a0).abs().max() / a0.abs().max()).item()


if __name__ == "__main__":
    run_miniapp("miniapp_gen_to_std", setup, run, flops, check)
