"""Adversarial-spectrum end-to-end eigensolver tests: clustered, repeated and
wide-dynamic-range eigenvalues exercise D&C deflation and the secular solver
through the full two-stage pipeline (reference analog: the eigensolver tests'
analytic element generators)."""
import numpy as np
import pytest
import torch

from dlaf_amd import Matrix, UpLo, hermitian_eigensolver
from dlaf_amd.matrix import util as mutil


def _from_spectrum(w, seed, dtype=torch.float64):
    n = w.shape[0]
    g = torch.Generator().manual_seed(seed)
    a = torch.randn(n, n, generator=g, dtype=torch.float64)
    if dtype.is_complex:
        a = a + 1j * torch.randn(n, n, generator=g, dtype=torch.float64)
    q, _ = torch.linalg.qr(a.to(dtype))
    return (q * w.to(dtype)) @ q.mH


def _run(w_true, seed, nb=64, dtype=torch.float64, tol=1e-10):
    n = w_true.shape[0]
    A = _from_spectrum(w_true, seed, dtype)
    A = 0.5 * (A + A.mH)
    mat = Matrix.create(n, n, nb, nb, dtype=dtype)
    mat.set_from_global(A)
    w, E = hermitian_eigensolver(UpLo.Lower, mat)
    Eg = E.to_global()
    scale = max(1.0, float(w_true.abs().max()))
    res = (A @ Eg - Eg @ torch.diag(w.to(dtype))).abs().max().item()
    orth = (Eg.mH @ Eg - torch.eye(n, dtype=dtype)).abs().max().item()
    assert res < tol * n * scale, f"res={res}"
    assert orth < tol * n, f"orth={orth}"
    err = np.abs(np.sort(w.numpy()) - np.sort(w_true.numpy())).max()
    assert err < tol * n * scale, f"eval err={err}"


def test_clustered_eigenvalues():
    """Tight clusters around a few centers (deflation-heavy merges)."""
    centers = torch.tensor([-3.0, 0.0, 5.0])
    w = torch.cat([c + 1e-9 * torch.arange(60, dtype=torch.float64)
                   for c in centers])
    _run(w, seed=11)


def test_repeated_eigenvalues():
    """Exactly repeated eigenvalues (full deflation of the repeats)."""
    w = torch.cat([torch.full((90,), 2.0), torch.full((90,), -1.0)]).double()
    _run(w, seed=12)


def test_wide_dynamic_range():
    """Eigenvalues spanning 12 orders of magnitude."""
    w = torch.logspace(-6, 6, 160, dtype=torch.float64)
    _run(w, seed=13, tol=1e-9)


def test_sign_symmetric_complex():
    """+/- paired spectrum, complex Hermitian."""
    half = torch.linspace(0.5, 4.0, 80, dtype=torch.float64)
    w = torch.cat([half, -half])
    _run(w, seed=14, dtype=torch.complex128)
