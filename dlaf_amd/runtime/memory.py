"""Pooled device and pinned-host buffers (the memory layer).

Counterpart of the reference's Umpire-backed ``MemoryChunk``/``MemoryView``
(``memory/memory_chunk.h:28-120``): reusable freelists keyed by
(bytes-bucket, device kind) over torch allocations. torch's caching
allocator already pools raw device memory; what this layer adds is

* PINNED host buffers for H2D/D2H staging (torch pins lazily and without
  reuse — each ``pin_memory()`` re-registers pages);
* workspace reuse ACROSS algorithm invocations (dinv/panel/scratch buffers
  are requested per call; the pool hands back the same storage instead of
  exercising the allocator on every solve);
* a bound: each freelist bucket keeps at most ``keep`` entries, so a
  long-lived process sweeping shapes cannot grow without limit (the
  round-1 ``_PLAN_CACHE`` finding).
"""

from __future__ import annotations

import threading
from typing import Dict, List, Tuple

import torch

_lock = threading.Lock()
_free: Dict[Tuple, List[torch.Tensor]] = {}
_KEEP = 8


def _bucket(nbytes: int) -> int:
    # power-of-two byte buckets >= 4 KiB
    b = 4096
    while b < nbytes:
        b <<= 1
    return b


def _key(nbytes: int, device: torch.device, pinned: bool) -> Tuple:
    return (_bucket(nbytes), device.type, device.index, pinned)


def acquire(shape, dtype: torch.dtype, device, pinned: bool = False
            ) -> torch.Tensor:
    """A zero-uninitialized buffer viewing pooled storage.

    ``pinned`` (CPU only): page-locked memory for async H2D/D2H.
    Return it with :func:`release` for reuse; dropping it is also safe
    (storage goes back to the torch allocator).
    """
    if not isinstance(device, torch.device):
        device = torch.device(device)
    numel = 1
    for s in shape:
        numel *= int(s)
    nbytes = numel * torch._utils._element_size(dtype)
    k = _key(nbytes, device, pinned)
    with _lock:
        lst = _free.get(k)
        raw = lst.pop() if lst else None
    if raw is None or raw.numel() * raw.element_size() < nbytes:
        nb = _bucket(nbytes)
        if pinned:
            # page-locking needs a GPU runtime; degrade gracefully on CPU
            pin_ok = torch.cuda.is_available()
            raw = torch.empty(nb, dtype=torch.uint8, pin_memory=pin_ok)
        else:
            raw = torch.empty(nb, dtype=torch.uint8, device=device)
    t = raw[:nbytes].view(dtype).view(*shape)
    t._pool_raw = raw  # keep the backing alive and identifiable
    return t


def release(t: torch.Tensor) -> None:
    raw = getattr(t, "_pool_raw", None)
    if raw is None:
        return
    pinned = raw.is_pinned() if raw.device.type == "cpu" else False
    k = _key(raw.numel(), raw.device, pinned)
    with _lock:
        lst = _free.setdefault(k, [])
        if len(lst) < _KEEP:
            lst.append(raw)


def pool_stats() -> Dict:
    with _lock:
        return {k: len(v) for k, v in _free.items()}


def clear() -> None:
    with _lock:
        _free.clear()
