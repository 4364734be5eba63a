#!/usr/bin/env python3
"""SYGV/HEGV miniapp (reference ``miniapp/miniapp_gen_eigensolver.cpp``)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from _harness import run_miniapp, random_herm, random_spd
from dlaf_amd import UpLo, hermitian_generalized_eigensolver


def setup(ctx):
    return {"a": random_herm(ctx), "b": random_spd(ctx)}


def run(ctx, st):
    return hermitian_generalized_eigensolver(UpLo.Lower, st["a"], st["b"], ctx.comm_grid)


if __name__ == "__main__":
    run_miniapp("miniapp_gen_eigensolver", setup, run, lambda ctx: None)
