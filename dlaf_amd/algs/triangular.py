"""Distributed triangular solve (TRSM) and multiplication (TRMM).

Counterpart of the reference's ``solver/triangular/impl.h`` and
``multiplication/triangular/impl.h``. Implemented incrementally; see tests.
"""
from __future__ import annotations


def triangular_solver(*args, **kwargs):
    raise NotImplementedError("triangular_solver: in progress")


def triangular_multiplication(*args, **kwargs):
    raise NotImplementedError("triangular_multiplication: in progress")
