"""Loader for the native HIP extension (dlaf_amd._hip).

Policy: on GPU the native kernels are THE compute path — if a CUDA tensor
reaches an op and the extension is missing, we raise (no silent eager
fallback). On CPU, torch ops are the reference backend (the analog of the
reference's Backend::MC).
"""

from __future__ import annotations

_ext = None
_tried = False


def get_ext():
    global _ext, _tried
    if _ext is None and not _tried:
        _tried = True
        try:
            from dlaf_amd import _hip  # type: ignore

            _ext = _hip
        except ImportError as e:  # pragma: no cover
            _ext = None
            _import_error = e
    if _ext is None:
        raise RuntimeError(
            "dlaf_amd._hip native extension not built. Build it in-tree with: "
            "python setup.py build_ext --inplace  (PYTORCH_ROCM_ARCH=gfx950)"
        )
    return _ext


def has_ext() -> bool:
    try:
        get_ext()
        return True
    except RuntimeError:
        return False
