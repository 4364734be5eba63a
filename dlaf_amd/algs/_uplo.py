"""Upper-triangle variants via the storage-transpose reduction.

The reference implements Upper natively (e.g. ``factorization/cholesky/
impl.h:317 call_U``); here Upper reduces to the Lower algorithm through the
identity U = L^H: conj-transpose the stored triangle, run Lower, conj-
transpose back. The two transposes are O(n^2) against the O(n^3) algorithms
(assembled via the replicated global matrix, which bounds this path to
single-node sizes — the same machine the framework targets).
"""
from __future__ import annotations

from ..matrix.matrix import Matrix


def transpose_storage(mat: Matrix) -> None:
    g = mat.to_global()
    mat.set_from_global(g.mH.contiguous())
