"""dlaf_amd — MI355X-native task-based distributed dense linear algebra.

A from-scratch rebuild of the capabilities of eth-cscs/DLA-Future,
architected for a single 8x AMD Instinct MI355X node:

* 2D block-cyclic tiled ``Matrix`` over a process grid (one process per GPU),
  ``torch.distributed`` with the RCCL backend over xGMI (``gloo`` on CPU for tests).
* Hand-written CDNA4 (gfx950) HIP kernels using fp64/fp32 MFMA with LDS staging for
  the hot per-tile ops (GEMM / SYRK / HERK / TRSM-by-block-inverse / POTRF / ...),
  fused across tiles so a whole trailing update is one kernel launch.
* HIP streams for panel lookahead and communication/compute overlap; comm issued on
  dedicated per-communicator streams in deterministic program order (the equivalent
  of the reference's exclusive ``CommunicatorPipeline`` ordering).

Public API (mirrors the reference's free-function API):
    cholesky_factorization, triangular_solver, triangular_multiplication,
    hermitian_multiplication, general_multiplication, inverse_from_cholesky_factor,
    triangular_inverse, generalized_to_standard, reduction_to_band,
    band_to_tridiagonal, tridiagonal_eigensolver, bt_band_to_tridiagonal,
    bt_reduction_to_band, hermitian_eigensolver, hermitian_generalized_eigensolver,
    max_norm
"""

from .types import Side, UpLo, Op, Diag  # noqa: F401
from .core.distribution import Distribution  # noqa: F401
from .comm.grid import CommGrid  # noqa: F401
from .matrix.matrix import Matrix
from .matrix.panel import Panel  # noqa: F401

from .algs.cholesky import cholesky_factorization  # noqa: F401
from .algs.triangular import triangular_solver, triangular_multiplication  # noqa: F401
from .algs.multiplication import (  # noqa: F401
    hermitian_multiplication,
    general_multiplication,
)
from .algs.inverse import triangular_inverse, inverse_from_cholesky_factor  # noqa: F401
from .algs.gen_to_std import generalized_to_standard  # noqa: F401
from .algs.norm import max_norm  # noqa: F401
from .algs.eigensolver import (  # noqa: F401
    hermitian_eigensolver,
    hermitian_generalized_eigensolver,
    get_band_size,
)
from .algs.red2band import reduction_to_band, bt_reduction_to_band  # noqa: F401
from .algs.band2tridiag import (  # noqa: F401
    band_to_tridiagonal,
    bt_band_to_tridiagonal,
)
from .algs.tridiag_dc import tridiagonal_eigensolver  # noqa: F401
from .algs.permutations import permute_columns, permute_rows  # noqa: F401
from .algs.redistribute import redistribute  # noqa: F401
from .matrix.mirror import MatrixMirror, MatrixRef, save_matrix, load_matrix  # noqa: F401
from .config import initialize, finalize, ScopedInitializer, get_tune_parameters  # noqa: F401
from .utils import Timer, trace_range  # noqa: F401

__version__ = "0.1.0"
