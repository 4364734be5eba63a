#!/usr/bin/env python3
"""SYGV/HEGV miniapp (reference ``miniapp/miniapp_gen_eigensolver.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_herm, random_spd
from dlaf_amd import UpLo, hermitian_generalized_eigensolver


def setup(ctx):
    st = {"a": random_herm(ctx), "b": random_spd(ctx)}
    if ctx.opts.check_result != "none":
        st["a0"], st["b0"] = st["a"].clone(), st["b"].clone()
    return st


def run(ctx, st):
    return hermitian_generalized_eigensolver(UpLo.Lower, st["a"], st["b"], ctx.comm_grid)


def check(ctx, st, result):
    """max |A E - B E diag(w)| / (|w|_max |B|_max)."""
    import torch
    w, evecs = result
    a = st["a0"].to_global()
    a = torch.tril(a) + torch.tril(a, -1).mH
    b = st["b0"].to_global()
    b = torch.tril(b) + torch.tril(b, -1).mH
    E = evecs.to_global()
    r = (a @ E - b @ (E * w.to(E.dtype))).abs().max()
    return (r / max(1.0, w.abs().max().item()) / b.abs().max()).item()


if __name__ == "__main__":
    run_miniapp("miniapp_gen_eigensolver", setup, run, lambda ctx: None, check)
