"""Full eigensolver pipeline correctness (local).

Mirrors ``test/unit/eigensolver/test_{eigensolver,gen_eigensolver}.cpp``:
residual ||A E - E diag(w)|| and orthogonality checks; partial spectrum;
generalized problem vs scipy reference.
"""

import numpy as np
import pytest
import scipy.linalg as sl
import torch

from dlaf_amd import (
    Matrix, UpLo, hermitian_eigensolver, hermitian_generalized_eigensolver,
)
from dlaf_amd.matrix import util as mutil


def _herm(a):
    return torch.tril(a) + torch.tril(a, -1).mH


def _make(n, nb, dtype, seed):
    mat = Matrix.create(n, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian(mat, seed=seed)
    return mat, _herm(mat.to_global())


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("n,nb,band", [(16, 4, 4), (32, 8, 4), (21, 7, 7), (64, 16, 8)])
def test_eigensolver_local(dtype, n, nb, band):
    mat, a0 = _make(n, nb, dtype, seed=51)
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat, band=band)
    E = evecs.to_global()
    scale = max(1.0, w.abs().max().item())
    res = (a0 @ E - E @ torch.diag(w.to(dtype))).abs().max().item()
    assert res < 1e-10 * n * scale, f"res={res}"
    orth = (E.mH @ E - torch.eye(n, dtype=dtype)).abs().max().item()
    assert orth < 1e-11 * n, f"orth={orth}"
    wref = np.linalg.eigvalsh(a0.numpy())
    werr = np.abs(np.sort(w.numpy()) - wref).max()
    assert werr < 1e-11 * n * scale, f"werr={werr}"


def test_eigensolver_partial_spectrum():
    n, nb = 32, 8
    mat, a0 = _make(n, nb, torch.float64, seed=53)
    ib, ie = 5, 20
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat, band=4,
                                     eigenvalues_index_begin=ib,
                                     eigenvalues_index_end=ie)
    assert w.shape[0] == ie - ib
    E = evecs.to_global()[:, : ie - ib]
    wref = np.linalg.eigvalsh(a0.numpy())
    assert np.abs(np.sort(w.numpy()) - wref[ib:ie]).max() < 1e-11 * n
    res = (a0 @ E - E @ torch.diag(w)).abs().max().item()
    assert res < 1e-10 * n, f"res={res}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_gen_eigensolver_local(dtype):
    n, nb = 24, 8
    A = Matrix.create(n, n, nb, nb, dtype=dtype)
    B = Matrix.create(n, n, nb, nb, dtype=dtype)
    mutil.set_random_hermitian(A, seed=61)
    mutil.set_random_hermitian_positive_definite(B, seed=62)
    a0, b0 = _herm(A.to_global()), _herm(B.to_global())
    w, evecs = hermitian_generalized_eigensolver(UpLo.Lower, A, B)
    E = evecs.to_global()
    # A E = B E diag(w)
    res = (a0 @ E - b0 @ E @ torch.diag(w.to(dtype))).abs().max().item()
    scale = max(1.0, w.abs().max().item())
    assert res < 1e-9 * n * scale, f"res={res}"
    wref = sl.eigh(a0.numpy(), b0.numpy(), eigvals_only=True)
    assert np.abs(np.sort(w.numpy()) - wref).max() < 1e-9 * n * scale
    # B-orthogonality
    borth = (E.mH @ b0 @ E - torch.eye(n, dtype=dtype)).abs().max().item()
    assert borth < 1e-9 * n, f"borth={borth}"


# ---------------- distributed (gloo) ----------------

from dlaf_amd import CommGrid  # noqa: E402
from dist_utils import run_distributed  # noqa: E402


def _dist_eig_worker(rank, ws, gr, gc, n, nb, dtype_str, gen):
    dtype = getattr(torch, dtype_str)
    grid = CommGrid(gr, gc)
    A = Matrix.create(n, n, nb, nb, dtype=dtype, grid=grid)
    mutil.set_random_hermitian(A, seed=71)
    a0 = _herm(A.to_global())
    if gen:
        B = Matrix.create(n, n, nb, nb, dtype=dtype, grid=grid)
        mutil.set_random_hermitian_positive_definite(B, seed=72)
        b0 = _herm(B.to_global())
        w, evecs = hermitian_generalized_eigensolver(UpLo.Lower, A, B, grid)
        E = evecs.to_global()
        res = (a0 @ E - b0 @ E @ torch.diag(w.to(dtype))).abs().max().item()
        wref = sl.eigh(a0.numpy(), b0.numpy(), eigvals_only=True)
    else:
        w, evecs = hermitian_eigensolver(UpLo.Lower, A, grid, band=nb)
        E = evecs.to_global()
        res = (a0 @ E - E @ torch.diag(w.to(dtype))).abs().max().item()
        wref = np.linalg.eigvalsh(a0.numpy())
    werr = np.abs(np.sort(w.numpy()) - wref).max()
    return res, werr


@pytest.mark.parametrize("gr,gc", [(1, 2), (2, 2)])
@pytest.mark.parametrize("dtype_str", ["float64", "complex128"])
def test_eigensolver_dist_cpu(gr, gc, dtype_str):
    n = 32
    out = run_distributed(_dist_eig_worker, gr * gc, args=(gr, gc, n, 8, dtype_str, False))
    for res, werr in out:
        assert res < 1e-9 * n, f"res={res}"
        assert werr < 1e-10 * n, f"werr={werr}"


def test_gen_eigensolver_dist_cpu():
    n = 24
    out = run_distributed(_dist_eig_worker, 2, args=(1, 2, n, 8, "float64", True))
    for res, werr in out:
        assert res < 1e-8 * n, f"res={res}"
        assert werr < 1e-9 * n, f"werr={werr}"


def _dist_partial_worker(rank, ws, gr, gc):
    grid = CommGrid(gr, gc)
    n, nb = 32, 8
    A = Matrix.create(n, n, nb, nb, dtype=torch.float64, grid=grid)
    mutil.set_random_hermitian(A, seed=91)
    a0 = _herm(A.to_global())
    ib, ie = 4, 20
    w, evecs = hermitian_eigensolver(UpLo.Lower, A, grid, band=8,
                                     eigenvalues_index_begin=ib,
                                     eigenvalues_index_end=ie)
    E = evecs.to_global()[:, : ie - ib]
    wref = np.linalg.eigvalsh(a0.numpy())
    werr = np.abs(np.sort(w.numpy()) - wref[ib:ie]).max()
    res = (a0 @ E - E @ torch.diag(w)).abs().max().item()
    return res, werr


def test_eigensolver_dist_partial_spectrum():
    for res, werr in run_distributed(_dist_partial_worker, 2, args=(1, 2)):
        assert res < 1e-9 * 32, f"res={res}"
        assert werr < 1e-10 * 32, f"werr={werr}"


def test_eigensolver_local_complex64():
    n, nb = 48, 16
    mat = Matrix.create(n, n, nb, nb, dtype=torch.complex64)
    mutil.set_random_hermitian(mat, seed=77)
    a0 = _herm(mat.to_global()).to(torch.complex128)
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat, band=8)
    E = evecs.to_global().to(torch.complex128)
    res = (a0 @ E - E @ torch.diag(w.to(torch.complex128))).abs().max().item()
    assert res < 1e-2, f"res={res}"
    wref = np.linalg.eigvalsh(a0.numpy())
    assert np.abs(np.sort(w.numpy().astype(np.float64)) - wref).max() < 1e-2


def test_eigensolver_deterministic():
    """Full-pipeline bitwise reproducibility (lockstep-replication contract)."""
    n, nb = 150, 64
    outs = []
    for _ in range(2):
        mat = Matrix.create(n, n, nb, nb, dtype=torch.float64)
        mutil.set_random_hermitian(mat, seed=9)
        w, E = hermitian_eigensolver(UpLo.Lower, mat)
        outs.append((w.clone(), E.to_global().clone()))
    assert torch.equal(outs[0][0], outs[1][0])
    assert torch.equal(outs[0][1], outs[1][1])


def test_eigensolver_band_not_dividing_nb():
    """nb % band != 0 must run the two-stage pipeline, not a dense-eigh
    cliff (round-1 verdict item 6): nb=96, band=40."""
    import torch
    from dlaf_amd import Matrix, UpLo, hermitian_eigensolver
    from dlaf_amd.matrix import util as mutil
    n, nb, band = 300, 96, 40
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cpu")
    mutil.set_random_hermitian(mat, seed=5)
    A = mat.to_global()
    A = torch.tril(A) + torch.tril(A, -1).mH
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat, band=band)
    E = evecs.to_global()
    R = A @ E - E @ torch.diag(w)
    assert float(R.abs().max()) < 1e-11 * n
    wref = torch.linalg.eigvalsh(A)
    assert float((w - wref).abs().max()) < 1e-11 * n
