"""HEMM / general GEMM / max_norm correctness vs torch reference.

Mirrors ``test/unit/multiplication/test_multiplication_{hermitian,general}.cpp``
and ``test/unit/auxiliary/mc/test_norm.cpp``.
"""

import pytest
import torch

from dlaf_amd import (
    Matrix, CommGrid, Side, UpLo, Op,
    hermitian_multiplication, general_multiplication, max_norm,
)
from dlaf_amd.matrix import util as mutil

from dist_utils import run_distributed


def _herm(a, uplo):
    if uplo == UpLo.Lower:
        return torch.tril(a) + torch.tril(a, -1).mH
    return torch.triu(a) + torch.triu(a, 1).mH


@pytest.mark.parametrize("side", [Side.Left, Side.Right])
@pytest.mark.parametrize("uplo", [UpLo.Lower, UpLo.Upper])
@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_hemm_local_cpu(side, uplo, dtype):
    m, n, nb = 17, 13, 5
    k = m if side == Side.Left else n
    A = Matrix.create(k, k, nb, nb, dtype=dtype)
    B = Matrix.create(m, n, nb, nb, dtype=dtype)
    C = Matrix.create(m, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian(A, seed=1)
    mutil.set_random(B, seed=2)
    mutil.set_random(C, seed=3)
    a, b, c0 = A.to_global(), B.to_global(), C.to_global()
    hermitian_multiplication(side, uplo, 0.75, A, B, 0.5, C)
    h = _herm(a, uplo)
    want = 0.75 * (h @ b if side == Side.Left else b @ h) + 0.5 * c0
    err = (C.to_global() - want).abs().max().item()
    assert err < 1e-11 * (m + n), f"err={err}"


def _dist_hemm_worker(rank, ws, gr, gc, side_s, uplo_s, dtype_str):
    dtype = getattr(torch, dtype_str)
    side, uplo = Side(side_s), UpLo(uplo_s)
    grid = CommGrid(gr, gc)
    m, n, nb = 22, 14, 4
    k = m if side == Side.Left else n
    A = Matrix.create(k, k, nb, nb, dtype=dtype, grid=grid)
    B = Matrix.create(m, n, nb, nb, dtype=dtype, grid=grid)
    C = Matrix.create(m, n, nb, nb, dtype=dtype, grid=grid)
    mutil.set_random_hermitian(A, seed=1)
    mutil.set_random(B, seed=2)
    mutil.set_random(C, seed=3)
    a, b, c0 = A.to_global(), B.to_global(), C.to_global()
    hermitian_multiplication(side, uplo, 1.0, A, B, -0.5, C, grid)
    h = _herm(a, uplo)
    want = (h @ b if side == Side.Left else b @ h) - 0.5 * c0
    return (C.to_global() - want).abs().max().item()


@pytest.mark.parametrize("side_s", ["L", "R"])
@pytest.mark.parametrize("uplo_s", ["L", "U"])
def test_hemm_dist_cpu(side_s, uplo_s):
    errs = run_distributed(_dist_hemm_worker, 4, args=(2, 2, side_s, uplo_s, "complex128"))
    for e in errs:
        assert e < 1e-10, f"err={e}"


@pytest.mark.parametrize("opA", [Op.NoTrans, Op.Trans, Op.ConjTrans])
@pytest.mark.parametrize("opB", [Op.NoTrans, Op.Trans])
def test_general_local_cpu(opA, opB):
    dtype = torch.complex128
    m, n, kk, nb = 12, 9, 15, 4
    sa = (m, kk) if opA is Op.NoTrans else (kk, m)
    sb = (kk, n) if opB is Op.NoTrans else (n, kk)
    A = Matrix.create(*sa, nb, nb, dtype=dtype)
    B = Matrix.create(*sb, nb, nb, dtype=dtype)
    C = Matrix.create(m, n, nb, nb, dtype=dtype)
    mutil.set_random(A, seed=1)
    mutil.set_random(B, seed=2)
    mutil.set_random(C, seed=3)

    def t(x, op):
        return x if op is Op.NoTrans else (x.mT if op is Op.Trans else x.mH)

    want = 2.0 * t(A.to_global(), opA) @ t(B.to_global(), opB) + 1.0 * C.to_global()
    general_multiplication(opA, opB, 2.0, A, B, 1.0, C)
    err = (C.to_global() - want).abs().max().item()
    assert err < 1e-11 * (m + n + kk), f"err={err}"


def _dist_gemm_worker(rank, ws, gr, gc):
    grid = CommGrid(gr, gc)
    m, n, kk, nb = 18, 14, 10, 4
    A = Matrix.create(m, kk, nb, nb, dtype=torch.float64, grid=grid)
    B = Matrix.create(kk, n, nb, nb, dtype=torch.float64, grid=grid)
    C = Matrix.create(m, n, nb, nb, dtype=torch.float64, grid=grid)
    mutil.set_random(A, seed=1)
    mutil.set_random(B, seed=2)
    mutil.set_random(C, seed=3)
    want = A.to_global() @ B.to_global() + C.to_global()
    general_multiplication(Op.NoTrans, Op.NoTrans, 1.0, A, B, 1.0, C, grid)
    return (C.to_global() - want).abs().max().item()


@pytest.mark.parametrize("gr,gc", [(2, 2), (1, 3)])
def test_general_dist_cpu(gr, gc):
    errs = run_distributed(_dist_gemm_worker, gr * gc, args=(gr, gc))
    for e in errs:
        assert e < 1e-11, f"err={e}"


def test_max_norm_local():
    mat = Matrix.create(19, 19, 5, 5, dtype=torch.float64)
    mutil.set_random(mat, seed=4)
    a = mat.to_global()
    assert abs(max_norm(mat) - a.abs().max().item()) < 1e-14
    assert abs(max_norm(mat, UpLo.Lower) - torch.tril(a).abs().max().item()) < 1e-14


def _dist_norm_worker(rank, ws, gr, gc):
    grid = CommGrid(gr, gc)
    mat = Matrix.create(21, 21, 4, 4, dtype=torch.float64, grid=grid)
    mutil.set_random(mat, seed=4)
    a = mat.to_global()
    return (max_norm(mat, grid=grid), a.abs().max().item())


def test_max_norm_dist():
    for got, want in run_distributed(_dist_norm_worker, 2, args=(1, 2)):
        assert abs(got - want) < 1e-14
