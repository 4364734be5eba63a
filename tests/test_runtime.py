"""Runtime layers: pooled memory, tile DAG engine."""
import torch

from dlaf_amd.runtime import memory as mempool
from dlaf_amd.runtime.dag import TileDag, cholesky_dag
from dlaf_amd.matrix.matrix import Matrix
from dlaf_amd.matrix import util as mutil
from dlaf_amd.types import UpLo


def test_memory_pool_reuse_and_bound():
    mempool.clear()
    t = mempool.acquire((64, 64), torch.float64, "cpu")
    raw = t._pool_raw
    mempool.release(t)
    t2 = mempool.acquire((64, 64), torch.float64, "cpu")
    assert t2._pool_raw is raw  # reused
    mempool.release(t2)
    # bound: releasing many buffers keeps at most _KEEP per bucket
    buts = [mempool.acquire((64, 64), torch.float64, "cpu") for _ in range(20)]
    for b in buts:
        mempool.release(b)
    assert max(pool := mempool.pool_stats().values()) <= 8, pool


def test_pinned_acquire():
    t = mempool.acquire((128,), torch.float32, "cpu", pinned=True)
    assert t.is_pinned() or not torch.cuda.is_available()
    mempool.release(t)


def test_dag_chain_cpu_order():
    dag = TileDag(torch.device("cpu"))
    log = []
    dag.submit(lambda s: log.append("w1"), writes=["a"])
    dag.submit(lambda s: log.append("r1"), reads=["a"])
    dag.submit(lambda s: log.append("r2"), reads=["a"])
    dag.submit(lambda s: log.append("w2"), writes=["a"])
    dag.wait_all()
    assert log == ["w1", "r1", "r2", "w2"]


def test_cholesky_dag_matches_fused():
    n, nb = 160, 32
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cpu")
    mutil.set_random_hermitian_positive_definite(mat, seed=4)
    A = mat.to_global().clone()
    cholesky_dag(mat)
    L = torch.tril(mat.to_global())
    ref = torch.tril(A) + torch.tril(A, -1).mT
    err = (L @ L.mT - ref).abs().max().item()
    assert err < 1e-10 * n, err


def test_mirror_pinned_roundtrip():
    mat = Matrix.create(64, 64, 16, 16, dtype=torch.float64, device="cpu")
    mutil.set_random(mat, seed=2)
    from dlaf_amd.matrix.mirror import MatrixMirror
    with MatrixMirror(mat, mat.device) as m2:
        assert m2 is mat  # same-device no-op


import pytest


@pytest.mark.gpu
def test_cholesky_dag_gpu():
    """DAG engine on real streams/events: per-tile Cholesky on device."""
    n, nb = 1024, 128
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda")
    mutil.set_random_hermitian_positive_definite(mat, seed=6)
    A = mat.to_global().clone()
    cholesky_dag(mat)
    L = torch.tril(mat.to_global())
    ref = torch.tril(A) + torch.tril(A, -1).mT
    err = (L @ L.mT - ref).abs().max().item()
    assert err < 1e-9 * n, err


@pytest.mark.gpu
def test_mirror_pinned_h2d_roundtrip():
    from dlaf_amd.matrix.mirror import MatrixMirror
    mat = Matrix.create(256, 256, 64, 64, dtype=torch.float64, device="cpu")
    mutil.set_random(mat, seed=8)
    before = mat.to_global().clone()
    with MatrixMirror(mat, "cuda") as dev_mat:
        assert dev_mat.device.type == "cuda"
        dev_mat.storage.mul_(2.0)
    assert torch.equal(mat.to_global(), 2.0 * before)


def test_assert_levels(monkeypatch):
    from dlaf_amd.core import asserts
    monkeypatch.setenv("DLAF_ASSERT_LEVEL", "1")
    asserts.dlaf_assert(True)
    try:
        asserts.dlaf_assert(False, "boom")
        raise RuntimeError("not raised")
    except asserts.DlafAssertError as e:
        assert "boom" in str(e)
    asserts.dlaf_assert_moderate(False)  # level 1: not checked
    monkeypatch.setenv("DLAF_ASSERT_LEVEL", "0")
    asserts.dlaf_assert(False)  # disabled
    monkeypatch.setenv("DLAF_ASSERT_LEVEL", "3")
    flag = []
    asserts.dlaf_assert_heavy(lambda: flag.append(1) or True)
    assert flag  # evaluated at level 3


def test_cholesky_entry_assert():
    import pytest as _pytest
    from dlaf_amd import cholesky_factorization, UpLo
    from dlaf_amd.core.asserts import DlafAssertError
    m = Matrix.create(12, 16, 4, 4, dtype=torch.float64)
    with _pytest.raises(DlafAssertError):
        cholesky_factorization(UpLo.Lower, m)


def test_timer_and_trace_range():
    """utils.Timer laps/report and the rocTX trace_range scope (no-op on
    CPU builds) — reference common/timer.h + pika instrumentation analog."""
    from dlaf_amd.utils import Timer, trace_range
    t = Timer()
    with trace_range("unit-test-range"):
        x = sum(range(1000))
    assert x == 499500
    d1 = t.lap("phase1")
    assert d1 >= 0.0
    t.lap("phase2")
    rep = t.report()
    assert "phase1" in rep and "phase2" in rep and "total" in rep
    assert t.elapsed() >= 0.0
