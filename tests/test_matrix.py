"""Matrix storage, generators and padding-invariant tests."""

import pytest
import torch

from dlaf_amd.core.distribution import Distribution
from dlaf_amd.matrix.matrix import Matrix
from dlaf_amd.matrix import util as mutil

DTYPES = [torch.float32, torch.float64, torch.complex64, torch.complex128]


@pytest.mark.parametrize("m,n,mb,nb", [(16, 16, 4, 4), (10, 14, 4, 3), (7, 7, 8, 8)])
def test_global_roundtrip(m, n, mb, nb):
    a = torch.randn(m, n, dtype=torch.float64)
    mat = Matrix.create(m, n, mb, nb)
    mat.set_from_global(a)
    assert torch.equal(mat.to_global(), a)


@pytest.mark.parametrize("dtype", DTYPES)
def test_hermitian_generator(dtype):
    mat = Matrix.create(24, 24, 5, 5, dtype=dtype)
    mutil.set_random_hermitian(mat, seed=3)
    a = mat.to_global()
    assert torch.allclose(a, a.mH.conj().mH.conj())  # sanity
    assert torch.allclose(a, a.mH, atol=0)


@pytest.mark.parametrize("dtype", DTYPES)
def test_spd_generator(dtype):
    mat = Matrix.create(20, 20, 6, 6, dtype=dtype)
    mutil.set_random_hermitian_positive_definite(mat, seed=1)
    a = mat.to_global()
    ev = torch.linalg.eigvalsh(a)
    assert ev.min().item() > 0


def test_identity_pad_invariant():
    # 20x20 with 6x6 tiles: last tile is 2x2, padded region must be identity
    mat = Matrix.create(20, 20, 6, 6, dtype=torch.float64)
    mutil.set_random_hermitian_positive_definite(mat)
    t = mat.tile((3, 3))
    assert torch.equal(t[2:, :2], torch.zeros(4, 2, dtype=torch.float64))
    assert torch.equal(t[:2, 2:], torch.zeros(2, 4, dtype=torch.float64))
    assert torch.equal(t[2:, 2:], torch.eye(4, dtype=torch.float64))


def test_tile_offsets():
    mat = Matrix.create(16, 16, 4, 4)
    flat = mat.storage.reshape(-1)
    for t in mat.dist.iter_local_tiles_global():
        off = mat.tile_offset(t)
        assert torch.equal(flat[off : off + 16].reshape(4, 4), mat.tile(t))


def test_distributed_views_rankwise():
    # simulate 2x3 grid rank views without torch.distributed
    m = n = 30
    a = torch.randn(m, n, dtype=torch.float64)
    acc = torch.zeros_like(a)
    for rr in range(2):
        for rc in range(3):
            d = Distribution(m, n, 4, 4, 2, 3, rr, rc)
            mat = Matrix(d)
            mat.set_from_global(a)
            acc += mat.to_global()
    assert torch.equal(acc, a)
