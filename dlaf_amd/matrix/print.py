"""Matrix printers (numpy / csv formats).

Counterpart of the reference's ``matrix/print_numpy.h`` / ``print_csv.h`` /
``print_gpu.h``: emit a distributed matrix in a format that can be pasted
into numpy or a spreadsheet. Every rank prints only with the assembled global
matrix (tests/debug scale only, like the reference).
"""

from __future__ import annotations

import io

from .matrix import Matrix


def _fmt(x) -> str:
    if isinstance(x, complex):
        return f"complex({x.real!r},{x.imag!r})"
    return repr(float(x))


def print_numpy(mat: Matrix, symbol: str = "mat", file=None) -> str:
    """numpy-parsable dump: ``mat = np.array([[...], ...])``."""
    a = mat.to_global().cpu()
    buf = io.StringIO()
    buf.write(f"{symbol} = np.array([")
    for i in range(a.shape[0]):
        row = ", ".join(_fmt(complex(v) if a.is_complex() else v.item()) for v in a[i])
        buf.write(f"[{row}],")
    buf.write(f"]).reshape{tuple(a.shape)}\n")
    out = buf.getvalue()
    if file is not None:
        file.write(out)
    return out


def print_csv(mat: Matrix, file=None) -> str:
    a = mat.to_global().cpu()
    buf = io.StringIO()
    for i in range(a.shape[0]):
        buf.write(",".join(_fmt(complex(v) if a.is_complex() else v.item()) for v in a[i]))
        buf.write("\n")
    out = buf.getvalue()
    if file is not None:
        file.write(out)
    return out
