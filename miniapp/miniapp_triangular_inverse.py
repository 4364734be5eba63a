#!/usr/bin/env python3
"""TRTRI miniapp (reference ``miniapp/miniapp_triangular_inverse.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_spd
from dlaf_amd import UpLo, Diag, triangular_inverse
from dlaf_amd.types import total_ops


def setup(ctx):
    a = random_spd(ctx)
    st = {"a": a}
    if ctx.opts.check_result != "none":
        st["a0"] = a.clone()
    return st


def run(ctx, st):
    triangular_inverse(UpLo.Lower, Diag.NonUnit, st["a"], ctx.comm_grid)


def flops(ctx):
    n = float(ctx.opts.matrix_size)
    return total_ops(ctx.dtype, n**3 / 6, n**3 / 6)


def check(ctx, st, _):
    """max |tril(X) tril(L0) - I|."""
    import torch
    X = torch.tril(st["a"].to_global())
    L0 = torch.tril(st["a0"].to_global())
    n = X.shape[0]
    eye = torch.eye(n, dtype=X.dtype, device=X.device)
    return (X @ L0 - eye).abs().max().item()


if __name__ == "__main__":
    run_miniapp("miniapp_triangular_inverse", setup, run, flops, check)
