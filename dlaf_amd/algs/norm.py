"""Max-norm of a distributed matrix.

Counterpart of ``auxiliary/norm/mc.h`` (max-abs element + MAX reduce).
"""
from __future__ import annotations


def max_norm(*args, **kwargs):
    raise NotImplementedError("max_norm: in progress")
