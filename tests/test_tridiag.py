"""Direct D&C tridiagonal eigensolver tests (reference
``test/unit/eigensolver/test_tridiag_solver_local.cpp`` analog): deflation-
heavy spectra, split points (zero off-diagonal), degenerate sizes, leaf
boundary sizes."""
import numpy as np
import pytest
import scipy.linalg as sl
import torch

from dlaf_amd.algs.tridiag_dc import tridiagonal_eigensolver


def _check(d, e, atol_scale=1e-12):
    w, E = tridiagonal_eigensolver(d.clone(), e.clone(), device="cpu")
    n = d.shape[0]
    T = torch.diag(d) + torch.diag(e, 1) + torch.diag(e, -1)
    scale = max(1.0, float(d.abs().max()), float(e.abs().max()) if n > 1 else 0.0)
    res = (T @ E - E * w).abs().max().item()
    orth = (E.T @ E - torch.eye(n, dtype=E.dtype)).abs().max().item()
    assert res < atol_scale * n * scale, f"res={res}"
    assert orth < atol_scale * n, f"orth={orth}"
    wr = np.sort(sl.eigvalsh_tridiagonal(d.numpy(), e.numpy())) if n > 1 else d.numpy()
    assert np.abs(np.sort(w.numpy()) - wr).max() < 1e-11 * n * scale
    assert bool((w[1:] >= w[:-1]).all())


@pytest.mark.parametrize("n", [1, 2, 3, 63, 64, 65, 129, 300])
def test_tridiag_random(n):
    g = torch.Generator().manual_seed(n)
    d = torch.randn(n, generator=g, dtype=torch.float64)
    e = torch.randn(max(n - 1, 0), generator=g, dtype=torch.float64)
    _check(d, e)


def test_tridiag_split_zero_offdiag():
    """A zero e entry decouples the problem (reference split handling)."""
    g = torch.Generator().manual_seed(7)
    n = 200
    d = torch.randn(n, generator=g, dtype=torch.float64)
    e = torch.randn(n - 1, generator=g, dtype=torch.float64)
    e[50] = 0.0
    e[130] = 0.0
    _check(d, e)


def test_tridiag_full_deflation():
    """Identity-like input: every rank-1 component deflates."""
    n = 150
    d = torch.full((n,), 3.0, dtype=torch.float64)
    e = torch.zeros(n - 1, dtype=torch.float64)
    _check(d, e)


def test_tridiag_heavy_deflation():
    """Many repeated diagonal entries and tiny couplings -> deflation path."""
    n = 180
    d = torch.ones(n, dtype=torch.float64)
    d[::3] = 2.0
    e = torch.full((n - 1,), 1e-18, dtype=torch.float64)
    e[::2] = 1e-3
    _check(d, e)


def test_tridiag_wilkinson():
    """Wilkinson W21+: clustered pairs of eigenvalues."""
    m = 10
    d = torch.tensor([abs(i - m) for i in range(2 * m + 1)], dtype=torch.float64)
    e = torch.ones(2 * m, dtype=torch.float64)
    _check(d, e)


def test_tridiag_deterministic():
    """Bitwise reproducibility — the distributed design replicates D&C in
    rank-lockstep and relies on identical inputs producing identical bits."""
    g = torch.Generator().manual_seed(77)
    n = 300
    d = torch.randn(n, generator=g, dtype=torch.float64)
    e = torch.randn(n - 1, generator=g, dtype=torch.float64)
    w1, E1 = tridiagonal_eigensolver(d.clone(), e.clone(), device="cpu")
    w2, E2 = tridiagonal_eigensolver(d.clone(), e.clone(), device="cpu")
    assert torch.equal(w1, w2) and torch.equal(E1, E2)


def test_tridiag_rot_batch_matches_sequential(monkeypatch):
    """DLAF_DC_ROT_BATCH=1 (rounds of disjoint Givens pairs) must produce
    bitwise the same result as the default sequential apply: the greedy
    round scheduler only commutes provably disjoint rotations."""
    g = torch.Generator().manual_seed(5)
    n = 260
    d = torch.ones(n, dtype=torch.float64)
    d[::2] = 2.0                       # heavy pair deflation -> many rotations
    e = 1e-3 * torch.randn(n - 1, generator=g, dtype=torch.float64).abs()
    monkeypatch.delenv("DLAF_DC_ROT_BATCH", raising=False)
    w1, E1 = tridiagonal_eigensolver(d.clone(), e.clone(), device="cpu")
    monkeypatch.setenv("DLAF_DC_ROT_BATCH", "1")
    w2, E2 = tridiagonal_eigensolver(d.clone(), e.clone(), device="cpu")
    assert torch.equal(w1, w2)
    assert torch.equal(E1, E2)
