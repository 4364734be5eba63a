"""ScaLAPACK-style API tests (reference ``test/unit/c_api``)."""

import numpy as np
import pytest
import torch

from dlaf_amd import capi
from dlaf_amd.capi import DLAF_descriptor

from dist_utils import run_distributed


def _local_blockcyclic(a, mb, nb, gr, gc, pr, pc):
    """Local block-cyclic part of global a for rank (pr, pc)."""
    m, n = a.shape
    rows = [i for i in range(m) if (i // mb) % gr == pr]
    cols = [j for j in range(n) if (j // nb) % gc == pc]
    return a[np.ix_(rows, cols)].copy(), rows, cols


def test_capi_potrf_local():
    n, nb = 24, 8
    rng = np.random.default_rng(3)
    a = rng.standard_normal((n, n))
    a = a @ a.T + n * np.eye(n)
    loc = a.copy()
    ctx = capi.dlaf_create_grid(1, 1)
    info = capi.dlaf_cholesky_factorization(ctx, "L", loc, DLAF_descriptor(n, n, nb, nb))
    capi.dlaf_free_grid(ctx)
    assert info == 0
    want = np.linalg.cholesky(a)
    assert np.abs(np.tril(loc) - want).max() < 1e-11 * n


def test_capi_syevd_local():
    n, nb = 20, 5
    rng = np.random.default_rng(5)
    a = rng.standard_normal((n, n))
    a = (a + a.T) / 2
    loc = np.tril(a).copy()
    w = np.zeros(n)
    z = np.zeros((n, n))
    ctx = capi.dlaf_create_grid(1, 1)
    info = capi.pXsyevd(ctx, "L", n, loc, DLAF_descriptor(n, n, nb, nb), w, z,
                        DLAF_descriptor(n, n, nb, nb))
    capi.dlaf_free_grid(ctx)
    assert info == 0
    wref = np.linalg.eigvalsh(a)
    assert np.abs(np.sort(w) - wref).max() < 1e-11 * n
    res = np.abs(a @ z - z * w).max()
    assert res < 1e-10 * n


def _dist_capi_worker(rank, ws, gr, gc):
    n, nb = 24, 4
    rng = np.random.default_rng(7)
    a = rng.standard_normal((n, n))
    a = a @ a.T + n * np.eye(n)
    pr, pc = rank // gc, rank % gc
    loc, rows, cols = _local_blockcyclic(a, nb, nb, gr, gc, pr, pc)
    ctx = capi.dlaf_create_grid(gr, gc)
    info = capi.dlaf_pdpotrf(ctx, "L", n, loc, 1, 1, DLAF_descriptor(n, n, nb, nb))
    capi.dlaf_free_grid(ctx)
    want = np.linalg.cholesky(a)
    want_loc = want[np.ix_(rows, cols)]
    mask = np.zeros((n, n), dtype=bool)
    mask[np.tril_indices(n)] = True
    mloc = mask[np.ix_(rows, cols)]
    return float(np.abs((loc - want_loc)[mloc]).max())


def test_capi_potrf_dist():
    errs = run_distributed(_dist_capi_worker, 4, args=(2, 2))
    for e in errs:
        assert e < 1e-10, f"err={e}"


def test_potrf_info_non_spd():
    """Non-positive-definite input reports ScaLAPACK-style info > 0."""
    import numpy as np
    from dlaf_amd import capi
    n, nb = 64, 32
    ctx = capi.dlaf_create_grid(1, 1)
    a = -np.eye(n, order="F")  # negative definite
    info = capi.dlaf_cholesky_factorization(
        ctx, "L", a, capi.DLAF_descriptor(n, n, nb, nb, ld=n))
    assert info > 0
    capi.dlaf_free_grid(ctx)


def test_capi_scalapack_shims():
    """dlaf_pdpotrf / pdpotri / pdsyevd ScaLAPACK-suffixed entry points."""
    import numpy as np
    from dlaf_amd import capi
    n, nb = 48, 16
    ctx = capi.dlaf_create_grid(1, 1)
    desc = capi.DLAF_descriptor(n, n, nb, nb, ld=n)
    rng = np.random.default_rng(3)
    a0 = rng.standard_normal((n, n))
    a0 = (a0 + a0.T) / 2 + 2 * n * np.eye(n)
    a = np.asfortranarray(a0)
    info = capi.dlaf_pdpotrf(ctx, "L", n, a, 1, 1, desc)
    assert info == 0
    L = np.tril(a)
    assert np.abs(L @ L.T - a0).max() < 1e-10 * n
    info = capi.dlaf_pdpotri(ctx, "L", n, a, 1, 1, desc)
    assert info == 0
    x = np.tril(a) + np.tril(a, -1).T
    assert np.abs(a0 @ x - np.eye(n)).max() < 1e-8 * n
    a = np.asfortranarray(a0)
    w = np.zeros(n)
    z = np.asfortranarray(np.zeros((n, n)))
    info = capi.dlaf_pdsyevd(ctx, "L", n, a, desc, w, z, desc)
    assert info == 0
    res = np.abs(a0 @ z - z * w).max()
    assert res < 1e-10 * n * max(1.0, np.abs(w).max())
    capi.dlaf_free_grid(ctx)


def test_capi_partial_spectrum():
    """dlaf_hermitian_eigensolver partial-spectrum indices."""
    import numpy as np
    from dlaf_amd import capi
    n, nb, il, iu = 40, 16, 5, 20
    ctx = capi.dlaf_create_grid(1, 1)
    desc = capi.DLAF_descriptor(n, n, nb, nb, ld=n)
    rng = np.random.default_rng(5)
    a0 = rng.standard_normal((n, n))
    a0 = (a0 + a0.T) / 2
    a = np.asfortranarray(a0)
    w = np.zeros(n)
    z = np.asfortranarray(np.zeros((n, n)))
    info = capi.dlaf_hermitian_eigensolver(ctx, "L", a, desc, w, z, desc,
                                           il=il, iu=iu)
    assert info == 0
    wr = np.sort(np.linalg.eigvalsh(a0))
    assert np.abs(w[: iu - il] - wr[il:iu]).max() < 1e-10 * n
    for j in range(iu - il):
        r = np.abs(a0 @ z[:, j] - w[j] * z[:, j]).max()
        assert r < 1e-9 * n
    capi.dlaf_free_grid(ctx)


def test_capi_potrf_upper():
    """uplo='U' routed through the C-API entry points (advisor finding 5)."""
    n, nb = 24, 8
    rng = np.random.default_rng(9)
    a = rng.standard_normal((n, n))
    a = a @ a.T + n * np.eye(n)
    loc = a.copy()
    ctx = capi.dlaf_create_grid(1, 1)
    info = capi.dlaf_cholesky_factorization(
        ctx, "U", loc, DLAF_descriptor(n, n, nb, nb))
    assert info == 0
    U = np.triu(loc)
    assert np.abs(U.T @ U - a).max() < 1e-11 * n
    # POTRI from the Upper factor
    info = capi.dlaf_inverse_from_cholesky_factor(
        ctx, "U", loc, DLAF_descriptor(n, n, nb, nb))
    capi.dlaf_free_grid(ctx)
    assert info == 0
    x = np.triu(loc) + np.triu(loc, 1).T
    assert np.abs(a @ x - np.eye(n)).max() < 1e-9 * n


def test_capi_heevd_upper():
    n, nb = 32, 8
    rng = np.random.default_rng(10)
    a = rng.standard_normal((n, n)) + 1j * rng.standard_normal((n, n))
    a = (a + a.conj().T) / 2
    loc = np.asfortranarray(a.copy())
    w = np.zeros(n)
    z = np.zeros((n, n), dtype=np.complex128, order="F")
    ctx = capi.dlaf_create_grid(1, 1)
    info = capi.dlaf_hermitian_eigensolver(
        ctx, "U", loc, DLAF_descriptor(n, n, nb, nb), w, z,
        DLAF_descriptor(n, n, nb, nb))
    capi.dlaf_free_grid(ctx)
    assert info == 0
    wref = np.linalg.eigvalsh(a)
    assert np.abs(np.sort(w) - wref).max() < 1e-11 * n
    assert np.abs(a @ z - z @ np.diag(w)).max() < 1e-10 * n


def test_capi_scalapack_shims_complex_and_generalized():
    """pzheevd / pdsygvd / pdtrtri shim coverage (the reference's full
    dtype-suffixed surface)."""
    n, nb = 32, 8
    ctx = capi.dlaf_create_grid(1, 1)
    desc = capi.DLAF_descriptor(n, n, nb, nb, ld=n)
    rng = np.random.default_rng(11)
    # pzheevd
    a0 = rng.standard_normal((n, n)) + 1j * rng.standard_normal((n, n))
    a0 = (a0 + a0.conj().T) / 2
    a = np.asfortranarray(a0)
    w = np.zeros(n)
    z = np.asfortranarray(np.zeros((n, n), dtype=np.complex128))
    info = capi.dlaf_pzheevd(ctx, "L", n, a, desc, w, z, desc)
    assert info == 0
    assert np.abs(a0 @ z - z * w).max() < 1e-10 * n * max(1.0, np.abs(w).max())
    # pdsygvd
    a0 = rng.standard_normal((n, n))
    a0 = (a0 + a0.T) / 2
    b0 = rng.standard_normal((n, n))
    b0 = b0 @ b0.T + n * np.eye(n)
    a = np.asfortranarray(a0)
    b = np.asfortranarray(b0)
    z = np.asfortranarray(np.zeros((n, n)))
    info = capi.dlaf_pdsygvd(ctx, "L", n, a, desc, b, desc, w, z, desc)
    assert info == 0
    res = np.abs(a0 @ z - b0 @ z * w).max()
    assert res < 1e-9 * n * max(1.0, np.abs(w).max())
    # pdtrtri on a well-conditioned lower factor
    l0 = np.tril(rng.standard_normal((n, n))) + 2 * n * np.eye(n)
    a = np.asfortranarray(l0.copy())
    info = capi.dlaf_pdtrtri(ctx, "L", "N", n, a, 1, 1, desc)
    assert info == 0
    assert np.abs(np.tril(a) @ l0 - np.eye(n)).max() < 1e-10 * n
    capi.dlaf_free_grid(ctx)


def test_capi_local_shape():
    """dlaf_local_shape: rank-local block-cyclic dims used by the C ABI."""
    ctx = capi.dlaf_create_grid(1, 1)
    lm, ln = capi.dlaf_local_shape(ctx, DLAF_descriptor(100, 70, 32, 16))
    capi.dlaf_free_grid(ctx)
    assert (lm, ln) == (100, 70)  # 1x1 grid: local == global
