"""MatrixMirror / MatrixRef / printers / config / save-load tests
(reference test/unit/matrix + test_init)."""

import os
import tempfile

import torch

from dlaf_amd import Matrix, UpLo
from dlaf_amd.matrix import util as mutil
from dlaf_amd.matrix.mirror import MatrixMirror, MatrixRef, save_matrix, load_matrix
from dlaf_amd.matrix.print import print_numpy, print_csv
from dlaf_amd import config


def test_matrix_mirror_same_device():
    m = Matrix.create(8, 8, 4, 4)
    mutil.set_random(m, seed=1)
    with MatrixMirror(m, "cpu") as t:
        assert t is m


def test_matrix_ref_view():
    m = Matrix.create(16, 16, 4, 4)
    mutil.set_random(m, seed=2)
    ref = MatrixRef(m, (4, 8), (8, 8))
    g = m.to_global()
    assert torch.equal(ref.to_global(), g[4:12, 8:16])
    assert torch.equal(ref.tile((0, 0)), m.tile((1, 2)))
    assert ref.dist.size == (8, 8)


def test_print_formats():
    m = Matrix.create(3, 3, 2, 2)
    mutil.set_random(m, seed=3)
    s = print_numpy(m, "a")
    import numpy as np
    ns = {"np": np}
    exec(s, ns)
    assert np.abs(ns["a"] - m.to_global().numpy()).max() < 1e-15
    csv = print_csv(m)
    assert len(csv.strip().split("\n")) == 3


def test_save_load_roundtrip():
    m = Matrix.create(10, 10, 4, 4)
    mutil.set_random(m, seed=4)
    m2 = Matrix.create(10, 10, 4, 4)
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "m.pt")
        save_matrix(m, p)
        load_matrix(p, m2)
    assert torch.equal(m.to_global(), m2.to_global())


def test_config_env_and_cli(monkeypatch):
    monkeypatch.setenv("DLAF_NUM_NP_GPU_STREAMS", "7")
    cfg = config.initialize(["--dlaf:num-hp-gpu-streams=5",
                             "--dlaf:eigensolver-min-band=64"])
    assert cfg.num_np_gpu_streams == 7      # env wins
    assert cfg.num_hp_gpu_streams == 5      # cli
    assert cfg.tune.eigensolver_min_band == 64
    with config.ScopedInitializer() as c2:
        assert c2.num_np_gpu_streams == 7
    config.finalize()


def test_set_random_hermitian_banded():
    """Banded generator (Appendix A surface): entries outside the band are
    zero, matrix is Hermitian."""
    import torch
    from dlaf_amd import Matrix
    from dlaf_amd.matrix import util as mutil
    n, nb, band = 48, 16, 8
    m = Matrix.create(n, n, nb, nb, dtype=torch.complex128)
    mutil.set_random_hermitian_banded(m, band, seed=3)
    a = m.to_global()
    full = torch.tril(a) + torch.tril(a, -1).mH
    assert torch.equal(torch.tril(full, -band - 1),
                       torch.zeros_like(full).tril(-band - 1))
    assert (full - full.mH).abs().max() == 0
    assert full.abs().max() > 0


def test_print_config(capsys):
    from dlaf_amd import config
    cfg = config.Configuration(print_config=True)
    config.initialize(user_cfg=cfg)
    out = capsys.readouterr().out
    assert "configuration" in out
    config.finalize()
