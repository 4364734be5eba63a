"""HIP stream / event runtime.

The MI355X-native replacement of the reference's pika runtime + stream/handle pools
(``src/init.cpp:94-146``): algorithms here are written as explicit stream programs —
a high-priority stream for the critical path (panel factorization + its broadcast),
a pool of normal-priority streams for trailing updates, and one dedicated stream per
communicator direction so collectives are issued in deterministic program order (the
role of the reference's ``CommunicatorPipeline::exclusive()``).

On CPU (the test environment) every stream is a no-op context and execution is
synchronous — the same algorithm code runs unchanged.
"""

from __future__ import annotations

import contextlib
from typing import List, Optional

import torch


class _NullStream:
    """CPU stand-in for torch.cuda.Stream."""

    def wait_stream(self, other):  # noqa: ANN001
        pass

    def wait_event(self, ev):  # noqa: ANN001
        pass

    def record_event(self, ev=None):  # noqa: ANN001
        return ev

    def synchronize(self):
        pass


class _NullEvent:
    def record(self, stream=None):  # noqa: ANN001
        pass

    def wait(self, stream=None):  # noqa: ANN001
        pass

    def synchronize(self):
        pass


class Runtime:
    """Per-process stream pools for one device."""

    def __init__(self, device: torch.device, n_hp: int = 2, n_np: int = 4):
        self.device = device
        self.gpu = device.type == "cuda"
        if self.gpu:
            with torch.cuda.device(device):
                self.hp_streams: List = [torch.cuda.Stream(priority=-1) for _ in range(n_hp)]
                self.np_streams: List = [torch.cuda.Stream(priority=0) for _ in range(n_np)]
                self.comm_stream = torch.cuda.Stream(priority=-1)
                self.d2h_stream = torch.cuda.Stream(priority=0)
        else:
            self.hp_streams = [_NullStream() for _ in range(n_hp)]
            self.np_streams = [_NullStream() for _ in range(n_np)]
            self.comm_stream = _NullStream()
            self.d2h_stream = _NullStream()
        self._rr_np = 0
        self._rr_hp = 0

    # round-robin pick (the reference round-robins 32+32 streams; a handful is
    # enough here because our kernels are fused across tiles)
    def np_stream(self):
        s = self.np_streams[self._rr_np % len(self.np_streams)]
        self._rr_np += 1
        return s

    def hp_stream(self):
        s = self.hp_streams[self._rr_hp % len(self.hp_streams)]
        self._rr_hp += 1
        return s

    def stream_ctx(self, stream):
        if self.gpu:
            return torch.cuda.stream(stream)
        return contextlib.nullcontext()

    def event(self):
        if self.gpu:
            return torch.cuda.Event()
        return _NullEvent()

    def synchronize(self):
        if self.gpu:
            torch.cuda.synchronize(self.device)

    def default_stream(self):
        if self.gpu:
            return torch.cuda.current_stream(self.device)
        return _NullStream()


_RUNTIME: Optional[Runtime] = None


def get_runtime(device: Optional[torch.device] = None) -> Runtime:
    """Process-global runtime. First call fixes the device."""
    global _RUNTIME
    if _RUNTIME is None:
        if device is None:
            device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        _RUNTIME = Runtime(device)
    return _RUNTIME


def reset_runtime():
    global _RUNTIME
    _RUNTIME = None
