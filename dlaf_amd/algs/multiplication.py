"""Hermitian and general tiled multiplication.

Counterpart of ``multiplication/hermitian/impl.h`` and
``multiplication/general/impl.h`` (the D&C eigensolver's workhorse GEMM).
"""
from __future__ import annotations


def hermitian_multiplication(*args, **kwargs):
    raise NotImplementedError("hermitian_multiplication: in progress")


def general_multiplication(*args, **kwargs):
    raise NotImplementedError("general_multiplication: in progress")
