"""Symmetric/Hermitian (generalized) eigensolver pipeline.

Counterpart of ``eigensolver/eigensolver/impl.h`` (red2band -> band2tridiag ->
tridiag D&C -> back-transforms) and ``eigensolver/gen_eigensolver/impl.h``.
"""
from __future__ import annotations


def hermitian_eigensolver(*args, **kwargs):
    raise NotImplementedError("hermitian_eigensolver: in progress")


def hermitian_generalized_eigensolver(*args, **kwargs):
    raise NotImplementedError("hermitian_generalized_eigensolver: in progress")
