"""Tracing and timing utilities.

Counterpart of the reference's observability story: the
reference relies on pika instrumentation + external tools and a wall-clock
``common/timer.h``; here ranges are emitted through the ROCm profiler marker
API (torch.cuda.nvtx maps to rocTX on ROCm builds, visible in rocprofv3
``--marker-trace``) plus a simple scoped wall-clock timer.
"""

from __future__ import annotations

import contextlib
import time
from typing import Dict, Optional

import torch


@contextlib.contextmanager
def trace_range(name: str):
    """Scoped rocTX/NVTX range (no-op overhead when no profiler attached)."""
    try:
        torch.cuda.nvtx.range_push(name)
        pushed = True
    except Exception:
        pushed = False
    try:
        yield
    finally:
        if pushed:
            try:
                torch.cuda.nvtx.range_pop()
            except Exception:
                pass


class Timer:
    """Wall-clock timer with named laps (reference ``common/timer.h``)."""

    def __init__(self, sync_device: Optional[torch.device] = None):
        self.sync_device = sync_device
        self.laps: Dict[str, float] = {}
        self._t0 = self._now()

    def _now(self) -> float:
        if self.sync_device is not None and self.sync_device.type == "cuda":
            torch.cuda.synchronize(self.sync_device)
        return time.perf_counter()

    def lap(self, name: str) -> float:
        t = self._now()
        dt = t - self._t0
        self.laps[name] = self.laps.get(name, 0.0) + dt
        self._t0 = t
        return dt

    def elapsed(self) -> float:
        return self._now() - self._t0

    def report(self) -> str:
        total = sum(self.laps.values())
        lines = [f"{k:24s} {v:9.3f}s ({100*v/total:5.1f}%)" for k, v in self.laps.items()]
        return "\n".join(lines + [f"{'total':24s} {total:9.3f}s"])
