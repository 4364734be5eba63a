"""Generalized-to-standard eigenproblem transform (HEGST, itype=1).

Counterpart of ``eigensolver/gen_to_std/impl.h``.
"""
from __future__ import annotations


def generalized_to_standard(*args, **kwargs):
    raise NotImplementedError("generalized_to_standard: in progress")
