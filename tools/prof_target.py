#!/usr/bin/env python3
"""Profiling targets for rocprofv3 runs: python tools/prof_target.py {gemm,potrf,chol}."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from dlaf_amd.ops import tile_ops as ops
from dlaf_amd.types import Op


def run_gemm(ntiles=1024, nb=512, iters=5):
    A = torch.randn(ntiles, nb, nb, dtype=torch.float64, device="cuda")
    C = torch.zeros_like(A)
    ts = nb * nb
    offs = [i * ts for i in range(ntiles)]
    descs = torch.from_numpy(ops.make_descs(offs, offs, offs)).to("cuda")
    for _ in range(iters):
        ops.gemm_fused(C, A, A, descs, nb, nb, nb, nb, nb, nb, Op.NoTrans, Op.Trans, -1.0, 1.0)
    torch.cuda.synchronize()


def run_potrf(nb=512, iters=20):
    a = torch.randn(nb, nb, dtype=torch.float64, device="cuda")
    a = a @ a.mH + nb * torch.eye(nb, dtype=torch.float64, device="cuda")
    t = a.clone()
    dinv = ops.dinv_workspace(nb, torch.float64, "cuda")
    for _ in range(iters):
        t.copy_(a)
        ops.potrf_tile(t, dinv)
    torch.cuda.synchronize()


def run_chol(n=8192, nb=512, iters=2):
    from dlaf_amd import Matrix, UpLo, cholesky_factorization
    from dlaf_amd.matrix import util as mutil

    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=1)
    pristine = mat.storage.clone()
    for _ in range(iters):
        mat.storage.copy_(pristine)
        cholesky_factorization(UpLo.Lower, mat)
    torch.cuda.synchronize()


if __name__ == "__main__":
    tgt = sys.argv[1] if len(sys.argv) > 1 else "gemm"
    {"gemm": run_gemm, "potrf": run_potrf, "chol": run_chol}[tgt]()
    print(f"done: {tgt}")
