"""Matrix generators — element-wise, rank-independent, device-vectorized.

Counterpart of the reference's analytic element setters
(``include/dlaf/util_matrix.h:236-551``: set / set_random /
set_random_hermitian[_positive_definite]). Like the reference, every generator is
a pure function of the GLOBAL element index, so each rank materializes its local
tiles without communication and all ranks agree on the matrix.

Randomness is a counter-based hash of (i, j, seed) evaluated vectorized on the
target device (values in [-1, 1]) — deterministic, reproducible, no generator
state.
"""

from __future__ import annotations

import math
from typing import Callable

import torch

from ..types import is_complex, real_dtype
from .matrix import Matrix


def set_elementwise(mat: Matrix, fn: Callable[[torch.Tensor, torch.Tensor], torch.Tensor]) -> None:
    """Fill ``mat`` with ``fn(I, J)`` where I, J are int64 global-index grids.

    Re-establishes the identity-extension padding invariant afterwards.
    """
    d = mat.dist
    dev = mat.device
    for li, lj in d.iter_local_tiles():
        gi, gj = d.global_tile_of_local((li, lj))
        ts = d.tile_size_of((gi, gj))
        rows = torch.arange(gi * d.mb, gi * d.mb + ts[0], device=dev, dtype=torch.int64)
        cols = torch.arange(gj * d.nb, gj * d.nb + ts[1], device=dev, dtype=torch.int64)
        I, J = torch.meshgrid(rows, cols, indexing="ij")
        tile = mat.storage[li, lj]
        tile.zero_()
        tile[: ts[0], : ts[1]] = fn(I, J).to(mat.dtype)
    mat._set_identity_pad()


def _hash01(I: torch.Tensor, J: torch.Tensor, salt: float, rdtype: torch.dtype) -> torch.Tensor:
    """Deterministic pseudo-random values in [-1, 1] from global indices."""
    x = I.to(rdtype) * 12.9898 + J.to(rdtype) * 78.2332 + salt
    v = torch.sin(x) * 43758.5453
    return (v - torch.floor(v)) * 2.0 - 1.0


def set_random(mat: Matrix, seed: int = 0) -> None:
    rd = real_dtype(mat.dtype)

    def fn(I, J):
        re = _hash01(I, J, 0.137 + seed, rd)
        if is_complex(mat.dtype):
            im = _hash01(I, J, 7.919 + seed, rd)
            return torch.complex(re, im)
        return re

    set_elementwise(mat, fn)


def _hermitian_fn(mat: Matrix, seed: int, diag_offset: float):
    rd = real_dtype(mat.dtype)

    def fn(I, J):
        lo = torch.minimum(I, J)
        hi = torch.maximum(I, J)
        re = _hash01(lo, hi, 0.137 + seed, rd)
        if is_complex(mat.dtype):
            im = _hash01(lo, hi, 7.919 + seed, rd)
            im = torch.where(I == J, torch.zeros_like(im), im)
            # conjugate below the diagonal so A[i,j] == conj(A[j,i])
            im = torch.where(I > J, -im, im)
            v = torch.complex(re, im)
            if diag_offset != 0.0:
                v = v + torch.where(
                    I == J, torch.full_like(re, diag_offset), torch.zeros_like(re)
                ).to(v.dtype)
            return v
        v = re
        if diag_offset != 0.0:
            v = v + torch.where(I == J, torch.full_like(re, diag_offset), torch.zeros_like(re))
        return v

    return fn


def set_random_hermitian(mat: Matrix, seed: int = 0) -> None:
    assert mat.dist.m == mat.dist.n and mat.dist.mb == mat.dist.nb
    set_elementwise(mat, _hermitian_fn(mat, seed, 0.0))


def set_random_hermitian_positive_definite(mat: Matrix, seed: int = 0) -> None:
    """Hermitian + diagonal offset 2n => strictly diagonally dominant => SPD/HPD.

    Same construction as the reference's
    ``set_random_hermitian_positive_definite`` (``util_matrix.h:529-531``).
    """
    assert mat.dist.m == mat.dist.n and mat.dist.mb == mat.dist.nb
    set_elementwise(mat, _hermitian_fn(mat, seed, 2.0 * mat.dist.n))


def set_identity(mat: Matrix, scale: float = 1.0) -> None:
    rd = real_dtype(mat.dtype)

    def fn(I, J):
        v = torch.where(I == J, torch.full(I.shape, scale, dtype=rd, device=I.device),
                        torch.zeros(I.shape, dtype=rd, device=I.device))
        return v

    set_elementwise(mat, fn)


def lower_triangle_dominant(mat: Matrix, seed: int = 0) -> None:
    """Random lower-triangular with dominant diagonal (well-conditioned TRSM input)."""
    rd = real_dtype(mat.dtype)
    n = mat.dist.n

    def fn(I, J):
        re = _hash01(I, J, 0.137 + seed, rd)
        if is_complex(mat.dtype):
            im = _hash01(I, J, 7.919 + seed, rd)
            v = torch.complex(re, im)
        else:
            v = re
        v = torch.where(I >= J, v, torch.zeros_like(v))
        diag = torch.where(I == J, torch.full_like(re, 2.0 * math.sqrt(n)), torch.zeros_like(re))
        return v + diag.to(v.dtype)

    set_elementwise(mat, fn)


def set_random_hermitian_banded(mat: Matrix, band: int, seed: int = 0) -> None:
    """Random Hermitian with zero entries outside |i-j| <= band (reference
    ``util_matrix.h`` set_random_hermitian_banded, band2tridiag test input)."""
    assert mat.dist.m == mat.dist.n and mat.dist.mb == mat.dist.nb
    base = _hermitian_fn(mat, seed, 0.0)

    def fn(I, J):
        v = base(I, J)
        mask = (I - J).abs() <= band
        return torch.where(mask, v, torch.zeros_like(v))

    set_elementwise(mat, fn)
