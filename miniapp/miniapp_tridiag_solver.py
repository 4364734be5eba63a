#!/usr/bin/env python3
"""tridiag D&C miniapp (reference ``miniapp/miniapp_tridiag_solver.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from _harness import run_miniapp
from dlaf_amd.algs.tridiag_dc import tridiagonal_eigensolver


def setup(ctx):
    n = ctx.opts.matrix_size
    g = torch.Generator().manual_seed(0)
    return {"d": torch.randn(n, generator=g, dtype=torch.float64),
            "e": torch.randn(n - 1, generator=g, dtype=torch.float64)}


def run(ctx, st):
    return tridiagonal_eigensolver(st["d"], st["e"], device=ctx.device)


def check(ctx, st, result):
    """max |T E - E diag(w)| / max(1, |w|_max)."""
    w, E = result
    n = st["d"].shape[0]
    T = torch.diag(st["d"]) + torch.diag(st["e"], 1) + torch.diag(st["e"], -1)
    T = T.to(E.device)
    r = (T @ E - E * w.to(E.dtype).to(E.device)).abs().max()
    return (r / max(1.0, w.abs().max().item())).item()


if __name__ == "__main__":
    run_miniapp("miniapp_tridiag_solver", setup, run, lambda ctx: None, check)
