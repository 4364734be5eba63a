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
