"""Triangular inverse (TRTRI) and inverse from Cholesky factor (POTRI).

Counterpart of ``inverse/triangular/impl.h`` and ``inverse/cholesky/impl.h``.
"""
from __future__ import annotations


def triangular_inverse(*args, **kwargs):
    raise NotImplementedError("triangular_inverse: in progress")


def inverse_from_cholesky_factor(*args, **kwargs):
    raise NotImplementedError("inverse_from_cholesky_factor: in progress")
