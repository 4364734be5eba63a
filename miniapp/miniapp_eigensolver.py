#!/usr/bin/env python3
"""SYEV/HEEV miniapp (reference ``miniapp/miniapp_eigensolver.cpp``):
time-to-solution of the full two-stage eigensolver."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from _harness import run_miniapp, random_herm
from dlaf_amd import UpLo, hermitian_eigensolver


def extra(p):
    p.add_argument("--band-size", type=int, default=0)


def setup(ctx):
    a = random_herm(ctx)
    return {"a": a, "ref": a.clone()}


def run(ctx, st):
    band = ctx.opts.band_size or None
    w, e = hermitian_eigensolver(UpLo.Lower, st["a"], ctx.comm_grid, band=band)
    return (w, e)


def check(ctx, st, result):
    w, evecs = result
    a = st["ref"].to_global()
    a = torch.tril(a) + torch.tril(a, -1).mH
    E = evecs.to_global()
    r = (a @ E - E @ torch.diag(w.to(E.dtype))).abs().max()
    return (r / max(1.0, w.abs().max())).item()


if __name__ == "__main__":
    run_miniapp("miniapp_eigensolver", setup, run, lambda ctx: None, check, extra=extra)
