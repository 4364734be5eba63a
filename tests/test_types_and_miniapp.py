"""Types/flop accounting + miniapp CLI contract tests."""
import subprocess
import sys
import os

import torch

from dlaf_amd.types import total_ops, dtype_char, real_dtype, is_complex


def test_total_ops_weights():
    # complex: 2 per add, 6 per mul (reference types.h:159-162)
    assert total_ops(torch.float64, 10, 20) == 30
    assert total_ops(torch.complex128, 10, 20) == 2 * 10 + 6 * 20
    assert dtype_char(torch.float32) == "s"
    assert dtype_char(torch.complex64) == "c"
    assert real_dtype(torch.complex128) is torch.float64
    assert is_complex(torch.complex64) and not is_complex(torch.float32)


def _run_miniapp(name, extra=()):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, os.path.join(root, "miniapp", name),
           "-m", "256", "-b", "64", "--nruns", "1", "--nwarmups", "0",
           "--backend", "mc", "--csv-output", *extra]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    return out.stdout


def test_miniapp_cholesky_csv_contract():
    out = _run_miniapp("miniapp_cholesky.py", ["--check-result", "last"])
    assert "CSVData-2" in out and "GFlops" in out
    assert "check residual" in out
    # human line format: [run] <t>s <gflops>GFlop/s ...
    assert "[0]" in out and "GFlop/s" in out


def test_miniapp_eigensolver_runs():
    out = _run_miniapp("miniapp_eigensolver.py", ["--check-result", "last"])
    assert "check residual" in out


def test_miniapp_trsm_runs():
    out = _run_miniapp("miniapp_triangular_solver.py")
    assert "CSVData-2" in out


import pytest

_ALL_MINIAPPS = [
    "miniapp_cholesky.py", "miniapp_triangular_solver.py",
    "miniapp_triangular_multiplication.py", "miniapp_gen_to_std.py",
    "miniapp_inverse_from_cholesky_factor.py", "miniapp_triangular_inverse.py",
    "miniapp_eigensolver.py", "miniapp_gen_eigensolver.py",
    "miniapp_reduction_to_band.py", "miniapp_band_to_tridiag.py",
    "miniapp_tridiag_solver.py", "miniapp_bt_band_to_tridiag.py",
    "miniapp_bt_reduction_to_band.py", "miniapp_communication.py",
    "miniapp_redistribution.py",
]


@pytest.mark.parametrize("app", _ALL_MINIAPPS)
def test_miniapp_smoke_all(app):
    """Every miniapp runs end-to-end with the shared CLI contract and emits
    the CSVData-2 machine row (reference miniapp/ inventory)."""
    extra = ["--check-result", "last"] if app not in (
        "miniapp_communication.py", "miniapp_redistribution.py") else []
    out = _run_miniapp(app, extra)
    if app == "miniapp_communication.py":
        # needs >1 rank for collectives; single-rank prints a notice (the
        # multi-rank path is covered by tests/test_comm.py)
        assert "CSVData-2" in out or "no communication" in out, out[-2000:]
    else:
        assert "CSVData-2" in out, out[-2000:]


def test_miniapp_cholesky_2rank():
    """Distributed miniapp harness path (grid 1x2, gloo) through torchrun —
    the reference miniapps run under MPI the same way."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29581",
           os.path.join(root, "miniapp", "miniapp_cholesky.py"),
           "-m", "256", "-b", "64", "--grid-rows", "1", "--grid-cols", "2",
           "--nruns", "1", "--nwarmups", "0", "--backend", "mc",
           "--csv-output", "--check-result", "last"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "CSVData-2" in out.stdout and "check residual" in out.stdout


def test_miniapp_complex_type():
    """--type z through the shared harness (reference dispatchMiniapp dtype
    axis)."""
    out = _run_miniapp("miniapp_eigensolver.py",
                       ["--type", "z", "--check-result", "last"])
    assert "CSVData-2" in out and "check residual" in out


def test_usage_doc_api_sequence():
    """The docs/USAGE.md Python-API sequence, shrunk to CPU sizes — keeps
    the documented entry points honest."""
    import torch
    from dlaf_amd import (Matrix, UpLo, Side, Op, Diag,
                          cholesky_factorization, triangular_solver,
                          hermitian_eigensolver)
    from dlaf_amd.matrix import util as mutil
    A = Matrix.create(96, 96, 32, 32, dtype=torch.float64, device="cpu")
    mutil.set_random_hermitian_positive_definite(A, seed=0)
    cholesky_factorization(UpLo.Lower, A)
    B = Matrix.create(96, 16, 32, 32, dtype=torch.float64, device="cpu")
    mutil.set_random(B, seed=1)
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit,
                      1.0, A, B)
    H = Matrix.create(80, 80, 32, 32, dtype=torch.float64, device="cpu")
    mutil.set_random_hermitian(H, seed=2)
    w, E = hermitian_eigensolver(UpLo.Lower, H)
    assert w.shape == (80,)
    H2 = Matrix.create(80, 80, 32, 32, dtype=torch.float64, device="cpu")
    mutil.set_random_hermitian(H2, seed=2)
    w10, E10 = hermitian_eigensolver(UpLo.Lower, H2,
                                     eigenvalues_index_begin=0,
                                     eigenvalues_index_end=10)
    assert w10.shape == (10,)
    assert torch.allclose(w10, w[:10], atol=1e-10)
