"""Multi-rank GPU code path smoke on ONE GPU: 2 processes share cuda:0 with
the gloo backend (gloo stages device tensors through the host), so the
distributed algorithms' stream/event schedules and collective ordering run
exactly as they will under RCCL — the only multi-rank GPU exercise possible
on a single-GPU box (RCCL refuses two ranks on one device)."""
import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from dist_utils import run_distributed

pytestmark = pytest.mark.gpu


def _worker_chol_trsm(rank, ws):
    from dlaf_amd import (Matrix, CommGrid, UpLo, Side, Op, Diag,
                          cholesky_factorization, triangular_solver)
    from dlaf_amd.matrix import util as mutil
    torch.cuda.set_device(0)
    n, nb = 768, 128
    grid = CommGrid(1, 2, device=torch.device("cuda", 0))
    A = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda", grid=grid)
    mutil.set_random_hermitian_positive_definite(A, seed=3)
    a0 = A.to_global().cpu()
    cholesky_factorization(UpLo.Lower, A, grid)
    torch.cuda.synchronize()
    L = torch.tril(A.to_global().cpu())
    err1 = (L @ L.mH - a0).abs().max().item()
    B = Matrix.create(n, 256, nb, nb, dtype=torch.float64, device="cuda", grid=grid)
    mutil.set_random(B, seed=4)
    b0 = B.to_global().cpu()
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0,
                      A, B, grid)
    torch.cuda.synchronize()
    err2 = (L @ B.to_global().cpu() - b0).abs().max().item()
    return max(err1, err2)


def _worker_eig(rank, ws):
    from dlaf_amd import Matrix, CommGrid, UpLo, hermitian_eigensolver
    from dlaf_amd.matrix import util as mutil
    torch.cuda.set_device(0)
    n, nb = 512, 128
    grid = CommGrid(2, 1, device=torch.device("cuda", 0))
    A = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda", grid=grid)
    mutil.set_random_hermitian(A, seed=5)
    a0 = A.to_global().cpu()
    a0 = torch.tril(a0) + torch.tril(a0, -1).mH
    w, E = hermitian_eigensolver(UpLo.Lower, A, grid)
    torch.cuda.synchronize()
    Eg = E.to_global().cpu()
    r = (a0 @ Eg - Eg @ torch.diag(w.cpu().to(Eg.dtype))).abs().max().item()
    return r


@pytest.mark.timeout(600)
def test_dist_gpu_cholesky_trsm_2ranks_one_gpu():
    errs = run_distributed(_worker_chol_trsm, 2)
    for e in errs:
        assert e < 1e-9, f"err={e}"


@pytest.mark.timeout(600)
def test_dist_gpu_eigensolver_2ranks_one_gpu():
    errs = run_distributed(_worker_eig, 2)
    for e in errs:
        assert e < 1e-9, f"err={e}"
