"""Scalar types, BLAS enums and flop accounting.

Counterpart of the reference's ``include/dlaf/types.h`` (SizeType/Device/Backend,
``total_ops`` flop weighting: complex = 2 flops per add, 6 per mul) and the blaspp
enum vocabulary used throughout ``include/dlaf/blas/tile.h``.
"""

from __future__ import annotations

import enum

import torch


class Side(enum.Enum):
    Left = "L"
    Right = "R"


class UpLo(enum.Enum):
    Lower = "L"
    Upper = "U"


class Op(enum.Enum):
    NoTrans = "N"
    Trans = "T"
    ConjTrans = "C"


class Diag(enum.Enum):
    Unit = "U"
    NonUnit = "N"


REAL_DTYPES = (torch.float32, torch.float64)
COMPLEX_DTYPES = (torch.complex64, torch.complex128)
ALL_DTYPES = REAL_DTYPES + COMPLEX_DTYPES

_DTYPE_CHAR = {
    torch.float32: "s",
    torch.float64: "d",
    torch.complex64: "c",
    torch.complex128: "z",
}

_REAL_OF = {
    torch.float32: torch.float32,
    torch.float64: torch.float64,
    torch.complex64: torch.float32,
    torch.complex128: torch.float64,
}


def dtype_char(dtype: torch.dtype) -> str:
    return _DTYPE_CHAR[dtype]


def is_complex(dtype: torch.dtype) -> bool:
    return dtype in COMPLEX_DTYPES


def real_dtype(dtype: torch.dtype) -> torch.dtype:
    return _REAL_OF[dtype]


def total_ops(dtype: torch.dtype, add: float, mul: float) -> float:
    """Weighted flop count: complex counts 2 per add and 6 per mul.

    Mirrors ``dlaf::total_ops`` (reference ``include/dlaf/types.h:159-162``) so the
    miniapp GFlop/s figures are comparable.
    """
    if is_complex(dtype):
        return 2.0 * add + 6.0 * mul
    return float(add + mul)
