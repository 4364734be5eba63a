"""ScaLAPACK-style drop-in API over local block-cyclic buffers.

Counterpart of the reference's C API (``include/dlaf_c/*``, ``src/c_api/*``):
grid management by integer context, ScaLAPACK-style descriptors, and
``p?potrf / p?potri / p?trtri / p?syevd / p?heevd / p?sygvd / p?hegvd``
entry points operating on the caller's LOCAL block-cyclic column-major
buffer (numpy or torch). Each call wraps the buffer into a tiled ``Matrix``
(mirrored to the GPU when available), runs the native algorithm, and copies
the result back — the flow of the reference's ``src/c_api/eigensolver/
eigensolver.h:30-73`` (host matrix -> MatrixMirror -> algorithm -> copy back).

Python is this framework's C-API surface (the package is the library); the
function names and argument conventions mirror the reference's so ScaLAPACK
callers can map 1:1.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

import numpy as np
import torch

from .types import UpLo, Diag
from .core.distribution import Distribution
from .comm.grid import CommGrid
from .matrix.matrix import Matrix
from .algs.cholesky import cholesky_factorization
from .algs.inverse import inverse_from_cholesky_factor, triangular_inverse
from .algs.eigensolver import hermitian_eigensolver, hermitian_generalized_eigensolver


@dataclass
class DLAF_descriptor:
    """ScaLAPACK-like descriptor (reference ``include/dlaf_c/desc.h``)."""
    m: int
    n: int
    mb: int
    nb: int
    isrc: int = 0
    jsrc: int = 0
    i: int = 1
    j: int = 1
    ld: int = 0


_grids: Dict[int, CommGrid] = {}
_next_ctx = [1]


def dlaf_create_grid(nprow: int, npcol: int, order: str = "R",
                     device: Optional[torch.device] = None) -> int:
    """Create a process grid context (reference ``dlaf_c/grid.h``).

    Row-major rank order only (the reference supports both; RCCL ranks here
    are torch.distributed ranks, which this framework orders row-major).

    Multi-process grids (nprow*npcol > 1): if torch.distributed is not yet
    initialized, it is initialized here from the torchrun/launcher
    environment (RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT) — RCCL when
    a GPU is visible, gloo otherwise. This is the C-ABI entry path for
    multi-rank callers (reference: ``src/c_api/grid.cpp`` builds the
    CommunicatorGrid from the caller's MPI_Comm; here the rendezvous is the
    launcher environment instead of MPI).
    """
    assert order.upper().startswith("R"), "row-major rank ordering only"
    import torch.distributed as tdist
    if (nprow * npcol > 1 and tdist.is_available()
            and not tdist.is_initialized()):
        import os
        assert "RANK" in os.environ and "WORLD_SIZE" in os.environ, (
            "multi-process grid: launch with torchrun-style env "
            "(RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT)")
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        tdist.init_process_group(backend=backend)
    grid = CommGrid(nprow, npcol, device=device)
    ctx = _next_ctx[0]
    _next_ctx[0] += 1
    _grids[ctx] = grid
    return ctx


def dlaf_local_shape(ctx: int, desc: "DLAF_descriptor"):
    """Rank-local (rows, cols) of the block-cyclic buffer described by
    ``desc`` on grid ``ctx`` — used by the C ABI (csrc/capi/dlaf_c.cpp) to
    wrap the caller's LOCAL panel with the right shape on >1x1 grids."""
    g = _grid(ctx)
    d = Distribution(desc.m, desc.n, desc.mb, desc.nb,
                     g.grid_rows, g.grid_cols, g.rank_row, g.rank_col,
                     desc.isrc, desc.jsrc)
    lm, ln = d.local_size
    return int(lm), int(ln)


def dlaf_free_grid(ctx: int) -> None:
    _grids.pop(ctx, None)


def _grid(ctx: int) -> CommGrid:
    return _grids[ctx]


def _device_for(grid: CommGrid):
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def _wrap_local(a_local, desc: DLAF_descriptor, grid: CommGrid, device) -> Matrix:
    """Tiled device Matrix from the caller's local block-cyclic buffer."""
    t = torch.as_tensor(a_local)
    assert desc.i == 1 and desc.j == 1, "sub-matrix offsets not supported"
    dist = Distribution(desc.m, desc.n, desc.mb, desc.nb,
                        grid.grid_rows, grid.grid_cols, grid.rank_row, grid.rank_col,
                        desc.isrc, desc.jsrc)
    mat = Matrix(dist, t.dtype, device, grid if grid.distributed else None)
    lr, lc = dist.local_nr_tiles
    for li in range(lr):
        for lj in range(lc):
            gi, gj = dist.global_tile_of_local((li, lj))
            tsr, tsc = dist.tile_size_of((gi, gj))
            blk = t[li * desc.mb: li * desc.mb + tsr, lj * desc.nb: lj * desc.nb + tsc]
            mat.storage[li, lj, :tsr, :tsc] = blk.to(device)
    mat._set_identity_pad()
    return mat


def _unwrap_local(mat: Matrix, a_local) -> None:
    t = torch.as_tensor(a_local)
    dist = mat.dist
    lr, lc = dist.local_nr_tiles
    for li in range(lr):
        for lj in range(lc):
            gi, gj = dist.global_tile_of_local((li, lj))
            tsr, tsc = dist.tile_size_of((gi, gj))
            blk = mat.storage[li, lj, :tsr, :tsc].to(t.device if t.is_cuda else "cpu")
            t[li * dist.mb: li * dist.mb + tsr, lj * dist.nb: lj * dist.nb + tsc] = blk
    if isinstance(a_local, np.ndarray):
        a_local[:] = t.numpy()


def _potrf_info(mat: Matrix, grid=None) -> int:
    """info > 0 when the input was not positive definite: a failed pivot
    sqrt produces NaN on that diagonal (reference counterpart: the
    cusolver-info device assert, ``src/cusolver/assert_info.cu:35``; here
    the check is an O(n) diagonal scan returned as ScaLAPACK-style info).

    Distributed grids: each rank scans only its owned diagonal tiles, then
    the smallest positive info is all-reduced over the full grid so EVERY
    rank returns the same global info (ScaLAPACK semantics — the reference
    returns a consistent info on all ranks)."""
    info = 0
    nt = mat.dist.nr_tiles[0]
    for k in range(nt):
        if mat.dist.rank_of_tile((k, k)) != (mat.dist.rank_row, mat.dist.rank_col):
            continue
        d = mat.tile((k, k)).diagonal()
        bad = torch.isnan(d.real if d.is_complex() else d)
        if bool(bad.any()):
            info = k * mat.dist.mb + int(bad.int().argmax()) + 1
            break
    if grid is not None and grid.distributed:
        import torch.distributed as tdist
        # min over positive infos == first failing pivot; encode 0 as +inf
        t = torch.tensor([float(info) if info > 0 else float("inf")],
                         dtype=torch.float64)
        tdist.all_reduce(t, op=tdist.ReduceOp.MIN, group=grid.full_group)
        v = float(t.item())
        info = 0 if v == float("inf") else int(v)
    return info


def dlaf_cholesky_factorization(ctx: int, uplo: str, a_local, desc: DLAF_descriptor) -> int:
    """``dlaf_cholesky_factorization_{s,d,c,z}`` analog; returns info (0 = ok,
    > 0 = leading minor of that order not positive definite)."""
    ul = UpLo.Upper if uplo.upper() == "U" else UpLo.Lower
    grid = _grid(ctx)
    dev = _device_for(grid)
    mat = _wrap_local(a_local, desc, grid, dev)
    cholesky_factorization(ul, mat, grid if grid.distributed else None)
    info = _potrf_info(mat, grid)
    _unwrap_local(mat, a_local)
    return info


def dlaf_inverse_from_cholesky_factor(ctx: int, uplo: str, a_local,
                                      desc: DLAF_descriptor) -> int:
    ul = UpLo.Upper if uplo.upper() == "U" else UpLo.Lower
    grid = _grid(ctx)
    mat = _wrap_local(a_local, desc, grid, _device_for(grid))
    inverse_from_cholesky_factor(ul, mat, grid if grid.distributed else None)
    _unwrap_local(mat, a_local)
    return 0


def dlaf_triangular_inverse(ctx: int, uplo: str, diag: str, a_local,
                            desc: DLAF_descriptor) -> int:
    ul = UpLo.Upper if uplo.upper() == "U" else UpLo.Lower
    grid = _grid(ctx)
    mat = _wrap_local(a_local, desc, grid, _device_for(grid))
    triangular_inverse(ul, Diag.Unit if diag.upper() == "U" else Diag.NonUnit,
                       mat, grid if grid.distributed else None)
    _unwrap_local(mat, a_local)
    return 0


def dlaf_hermitian_eigensolver(ctx: int, uplo: str, a_local, desc: DLAF_descriptor,
                               w_out, z_local, descz: DLAF_descriptor,
                               il: int = 0, iu: Optional[int] = None) -> int:
    """``dlaf_{symmetric,hermitian}_eigensolver[_partial_spectrum]`` analog.

    w_out: [n] real output buffer; z_local: local eigenvector buffer.
    """
    grid = _grid(ctx)
    dev = _device_for(grid)
    mat = _wrap_local(a_local, desc, grid, dev)
    if uplo.upper() == "U":
        # A is Hermitian: conj-transposing the storage turns the given upper
        # triangle into the lower triangle the pipeline reads.
        from .algs._uplo import transpose_storage
        transpose_storage(mat)
    w, evecs = hermitian_eigensolver(UpLo.Lower, mat, grid if grid.distributed else None,
                                     eigenvalues_index_begin=il,
                                     eigenvalues_index_end=iu)
    wt = torch.as_tensor(w_out)
    wt[: w.shape[0]] = w.cpu().to(wt.dtype)
    if isinstance(w_out, np.ndarray):
        w_out[: w.shape[0]] = wt[: w.shape[0]].numpy()
    _unwrap_local(evecs, z_local)
    _unwrap_local(mat, a_local)
    return 0


def dlaf_hermitian_generalized_eigensolver(ctx: int, uplo: str, a_local,
                                           desca: DLAF_descriptor, b_local,
                                           descb: DLAF_descriptor, w_out,
                                           z_local, descz: DLAF_descriptor,
                                           factorized: bool = False) -> int:
    grid = _grid(ctx)
    dev = _device_for(grid)
    mat_a = _wrap_local(a_local, desca, grid, dev)
    mat_b = _wrap_local(b_local, descb, grid, dev)
    if uplo.upper() == "U":
        from .algs._uplo import transpose_storage
        transpose_storage(mat_a)
        transpose_storage(mat_b)
    w, evecs = hermitian_generalized_eigensolver(
        UpLo.Lower, mat_a, mat_b, grid if grid.distributed else None,
        factorized=factorized)
    wt = torch.as_tensor(w_out)
    wt[: w.shape[0]] = w.cpu().to(wt.dtype)
    if isinstance(w_out, np.ndarray):
        w_out[: w.shape[0]] = wt[: w.shape[0]].numpy()
    _unwrap_local(evecs, z_local)
    _unwrap_local(mat_b, b_local)
    return 0


# ---- ScaLAPACK-style shims (reference dlaf_c/...: dlaf_p{s,d,c,z}potrf etc.) ----

def _sl_desc(n, mb, nb, uplo_n=None, m=None) -> DLAF_descriptor:
    return DLAF_descriptor(m if m is not None else n, n, mb, nb)


def pXpotrf(ctx: int, uplo: str, n: int, a_local, ia: int, ja: int,
            desca: DLAF_descriptor) -> int:
    assert (ia, ja) == (1, 1)
    return dlaf_cholesky_factorization(ctx, uplo, a_local, desca)


def pXpotri(ctx: int, uplo: str, n: int, a_local, ia: int, ja: int,
            desca: DLAF_descriptor) -> int:
    assert (ia, ja) == (1, 1)
    return dlaf_inverse_from_cholesky_factor(ctx, uplo, a_local, desca)


def pXtrtri(ctx: int, uplo: str, diag: str, n: int, a_local, ia: int, ja: int,
            desca: DLAF_descriptor) -> int:
    assert (ia, ja) == (1, 1)
    return dlaf_triangular_inverse(ctx, uplo, diag, a_local, desca)


def pXsyevd(ctx: int, uplo: str, n: int, a_local, desca: DLAF_descriptor,
            w_out, z_local, descz: DLAF_descriptor) -> int:
    return dlaf_hermitian_eigensolver(ctx, uplo, a_local, desca, w_out, z_local, descz)


def pXsygvd(ctx: int, uplo: str, n: int, a_local, desca, b_local, descb,
            w_out, z_local, descz) -> int:
    return dlaf_hermitian_generalized_eigensolver(ctx, uplo, a_local, desca,
                                                  b_local, descb, w_out, z_local, descz)


# dtype-suffixed aliases matching the reference's C symbol names
dlaf_pdpotrf = dlaf_pspotrf = dlaf_pcpotrf = dlaf_pzpotrf = pXpotrf
dlaf_pdpotri = dlaf_pspotri = dlaf_pcpotri = dlaf_pzpotri = pXpotri
dlaf_pdtrtri = dlaf_pstrtri = dlaf_pctrtri = dlaf_pztrtri = pXtrtri
dlaf_pdsyevd = dlaf_pssyevd = pXsyevd
dlaf_pcheevd = dlaf_pzheevd = pXsyevd
dlaf_pdsygvd = dlaf_pssygvd = pXsygvd
dlaf_pchegvd = dlaf_pzhegvd = pXsygvd
