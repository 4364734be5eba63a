"""band_to_tridiagonal + back-transform correctness.

Mirrors ``test/unit/eigensolver/test_band_to_tridiag.cpp`` strategy: the
tridiagonal must be similar to the band matrix (same eigenvalues), and
applying the back-transform to the tridiagonal eigenvectors must give
eigenvectors of the band matrix.
"""

import numpy as np
import pytest
import scipy.linalg as sl
import torch

from dlaf_amd import Matrix, UpLo
from dlaf_amd.matrix import util as mutil
from dlaf_amd.algs.band2tridiag import band_to_tridiagonal, bt_band_to_tridiagonal


def _band_matrix(n, b, dtype, seed=0):
    """Random Hermitian band matrix as a tiled Matrix (nb = b) + dense copy."""
    g = torch.Generator().manual_seed(seed)
    a = torch.randn(n, n, generator=g, dtype=torch.float64)
    if dtype.is_complex:
        a = a + 1j * torch.randn(n, n, generator=g, dtype=torch.float64)
    a = a.to(dtype)
    a = torch.tril(a)
    if dtype.is_complex:  # Hermitian: real diagonal
        a = a - torch.diag(torch.diagonal(a)) + torch.diag(torch.diagonal(a).real.to(dtype))
    mask = torch.ones(n, n).tril().triu(-b) > 0
    a = torch.where(torch.tril(mask), a, torch.zeros_like(a))
    a = torch.tril(a, 0)
    full = a + torch.tril(a, -1).mH
    mat = Matrix.create(n, n, b, b, dtype=dtype)
    mat.set_from_global(full)
    return mat, full


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("n,b", [(8, 2), (16, 4), (21, 4), (33, 8), (12, 12)])
def test_band2tridiag_eigenvalues(dtype, n, b):
    mat, full = _band_matrix(n, b, dtype, seed=41)
    tri = band_to_tridiagonal(UpLo.Lower, b, mat)
    w_t = np.sort(sl.eigvalsh_tridiagonal(tri.d.numpy(), tri.e.numpy()))
    w_f = np.sort(np.linalg.eigvalsh(full.numpy()))
    err = np.abs(w_t - w_f).max()
    scale = max(1.0, np.abs(w_f).max())
    assert err < 1e-12 * n * scale, f"err={err}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
@pytest.mark.parametrize("n,b", [(16, 4), (21, 4), (33, 8)])
def test_band2tridiag_backtransform(dtype, n, b):
    mat, full = _band_matrix(n, b, dtype, seed=43)
    tri = band_to_tridiagonal(UpLo.Lower, b, mat)
    w, v = sl.eigh_tridiagonal(tri.d.numpy(), tri.e.numpy())
    E = torch.from_numpy(v).to(dtype)
    bt_band_to_tridiagonal(E, tri)
    # A E = E diag(w)
    res = (full @ E - E @ torch.diag(torch.from_numpy(w).to(dtype))).abs().max().item()
    scale = max(1.0, np.abs(w).max())
    assert res < 1e-12 * n * scale, f"res={res}"
    orth = (E.mH @ E - torch.eye(n, dtype=dtype)).abs().max().item()
    assert orth < 1e-12 * n, f"orth={orth}"


@pytest.mark.parametrize("dtype", [torch.float64, torch.complex128])
def test_bt_window_merge_equivalence(dtype):
    """Merged block-WY application (window_merge m) is algebraically exact:
    identical output to the unmerged ascending-k chain."""
    from dlaf_amd.config import get_tune_parameters
    n, b, nE = 230, 8, 13
    mat, _ = _band_matrix(n, b, dtype, seed=45)
    tri = band_to_tridiagonal(UpLo.Lower, b, mat)
    tp = get_tune_parameters()
    outs = {}
    for m in (1, 3, 4):
        torch.manual_seed(7)
        E = torch.randn(n, nE, dtype=torch.float64).to(dtype)
        old = tp.bt_band_to_tridiag_window_merge
        tp.bt_band_to_tridiag_window_merge = m
        try:
            bt_band_to_tridiagonal(E, tri, group_size=16)
        finally:
            tp.bt_band_to_tridiag_window_merge = old
        outs[m] = E
    assert torch.allclose(outs[1], outs[4], atol=1e-12, rtol=1e-12)
    assert torch.allclose(outs[1], outs[3], atol=1e-12, rtol=1e-12)
