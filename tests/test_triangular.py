"""TRSM / TRMM correctness: all {side, uplo, op, diag} cases vs torch reference.

Mirrors the reference's ``test/unit/solver/test_triangular.cpp`` and
``test/unit/multiplication/test_multiplication_triangular.cpp``: typed over
dtypes x cases x {size, block} tables including degenerate and non-divisible
shapes; local CPU and distributed CPU (gloo).
"""

import pytest
import torch

from dlaf_amd import (
    Matrix, CommGrid, Side, UpLo, Op, Diag,
    triangular_solver, triangular_multiplication,
)
from dlaf_amd.matrix import util as mutil

from dist_utils import run_distributed

SIDES = [Side.Left, Side.Right]
UPLOS = [UpLo.Lower, UpLo.Upper]
OPS = [Op.NoTrans, Op.Trans, Op.ConjTrans]
DIAGS = [Diag.NonUnit, Diag.Unit]


def _t(x, op):
    return x if op is Op.NoTrans else (x.mT if op is Op.Trans else x.mH)


def _tri(a, uplo, diag):
    lower = uplo == UpLo.Lower
    t = torch.tril(a) if lower else torch.triu(a)
    if diag == Diag.Unit:
        n = a.shape[0]
        t = t - torch.diag(torch.diagonal(t)) + torch.eye(n, dtype=a.dtype)
    return t


def _make_ab(side, m, n, nb, dtype, grid=None, seed=3):
    k = m if side == Side.Left else n
    A = Matrix.create(k, k, nb, nb, dtype=dtype, grid=grid)
    B = Matrix.create(m, n, nb, nb, dtype=dtype, grid=grid)
    mutil.set_random_hermitian_positive_definite(A, seed=seed)  # well-conditioned
    mutil.set_random(B, seed=seed + 1)
    return A, B


def _ref_trsm(side, uplo, op, diag, alpha, a_full, b):
    tri = _t(_tri(a_full, uplo, diag), op)
    if side == Side.Left:
        return torch.linalg.solve(tri, alpha * b)
    return torch.linalg.solve(tri.mT, (alpha * b).mT).mT


def _ref_trmm(side, uplo, op, diag, alpha, a_full, b):
    tri = _t(_tri(a_full, uplo, diag), op)
    return alpha * (tri @ b) if side == Side.Left else alpha * (b @ tri)


def _tol(dtype, n):
    base = 2e-4 if dtype in (torch.float32, torch.complex64) else 5e-10
    return base * max(1, n)


@pytest.mark.parametrize("side", SIDES)
@pytest.mark.parametrize("uplo", UPLOS)
@pytest.mark.parametrize("op", OPS)
@pytest.mark.parametrize("diag", DIAGS)
def test_trsm_local_cpu_cases(side, uplo, op, diag):
    dtype = torch.complex128 if op is Op.ConjTrans else torch.float64
    m, n, nb = 17, 13, 5
    A, B = _make_ab(side, m, n, nb, dtype)
    a_full, b0 = A.to_global(), B.to_global()
    triangular_solver(side, uplo, op, diag, 1.5, A, B)
    want = _ref_trsm(side, uplo, op, diag, 1.5, a_full, b0)
    err = (B.to_global() - want).abs().max().item()
    assert err < _tol(dtype, m + n), f"err={err}"


@pytest.mark.parametrize("side", SIDES)
@pytest.mark.parametrize("uplo", UPLOS)
@pytest.mark.parametrize("op", OPS)
@pytest.mark.parametrize("diag", DIAGS)
def test_trmm_local_cpu_cases(side, uplo, op, diag):
    dtype = torch.complex128 if op is Op.ConjTrans else torch.float64
    m, n, nb = 17, 13, 5
    A, B = _make_ab(side, m, n, nb, dtype)
    a_full, b0 = A.to_global(), B.to_global()
    triangular_multiplication(side, uplo, op, diag, 0.5, A, B)
    want = _ref_trmm(side, uplo, op, diag, 0.5, a_full, b0)
    err = (B.to_global() - want).abs().max().item()
    assert err < _tol(dtype, m + n), f"err={err}"


@pytest.mark.parametrize("m,n,nb", [(0, 4, 4), (4, 0, 4), (4, 4, 8), (24, 16, 8)])
def test_trsm_local_cpu_shapes(m, n, nb):
    A, B = _make_ab(Side.Left, m, n, nb, torch.float64)
    a_full, b0 = A.to_global(), B.to_global()
    triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0, A, B)
    if m and n:
        want = _ref_trsm(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0, a_full, b0)
        err = (B.to_global() - want).abs().max().item()
        assert err < _tol(torch.float64, m + n), f"err={err}"


def _dist_worker(rank, world_size, gr, gc, which, side_s, uplo_s, op_s, m, n, nb, dtype_str):
    dtype = getattr(torch, dtype_str)
    side, uplo, op = Side(side_s), UpLo(uplo_s), Op(op_s)
    grid = CommGrid(gr, gc)
    A, B = _make_ab(side, m, n, nb, dtype, grid=grid)
    a_full, b0 = A.to_global(), B.to_global()
    if which == "trsm":
        triangular_solver(side, uplo, op, Diag.NonUnit, 1.0, A, B, grid)
        want = _ref_trsm(side, uplo, op, Diag.NonUnit, 1.0, a_full, b0)
    else:
        triangular_multiplication(side, uplo, op, Diag.NonUnit, 1.0, A, B, grid)
        want = _ref_trmm(side, uplo, op, Diag.NonUnit, 1.0, a_full, b0)
    return (B.to_global() - want).abs().max().item()


@pytest.mark.parametrize("which", ["trsm", "trmm"])
@pytest.mark.parametrize("side_s,uplo_s,op_s", [
    ("L", "L", "N"), ("L", "L", "T"), ("L", "U", "N"), ("L", "U", "T"),
    ("R", "L", "N"), ("R", "L", "T"), ("R", "U", "N"), ("R", "U", "T"),
])
def test_triangular_dist_cpu(which, side_s, uplo_s, op_s):
    errs = run_distributed(_dist_worker, 4,
                           args=(2, 2, which, side_s, uplo_s, op_s, 24, 20, 4, "float64"))
    for e in errs:
        assert e < _tol(torch.float64, 44), f"err={e}"


@pytest.mark.parametrize("which", ["trsm", "trmm"])
def test_triangular_dist_cpu_complex(which):
    errs = run_distributed(_dist_worker, 2,
                           args=(1, 2, which, "L", "L", "C", 18, 12, 5, "complex128"))
    for e in errs:
        assert e < _tol(torch.complex128, 30), f"err={e}"
