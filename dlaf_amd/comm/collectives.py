"""Thin stream-aware wrappers over torch.distributed collectives.

Counterpart of the reference's async tile collectives
(``communication/kernels/*`` — schedule_bcast_send/recv, schedule_all_reduce,
schedule_reduce_*, schedule_send/recv). Differences by design:

* RCCL is GPU-aware: no pinned-host staging (the reference's ``withTemporaryTile``
  CommDevice path vanishes).
* Ordering: callers issue collectives on a per-communicator HIP stream in
  deterministic program order (see ``CommGrid`` docstring); with ``gloo`` on CPU the
  calls are host-blocking and ordering is trivial.
* Complex tensors are viewed as real pairs (same storage) so every backend —
  including gloo — handles them; sums are elementwise so SUM semantics is preserved.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def _commview(t: torch.Tensor) -> torch.Tensor:
    if t.is_complex():
        return torch.view_as_real(t)
    return t


def broadcast(t: torch.Tensor, src_global_rank: int, group) -> None:
    dist.broadcast(_commview(t), src=src_global_rank, group=group)


def all_reduce_sum(t: torch.Tensor, group) -> None:
    dist.all_reduce(_commview(t), op=dist.ReduceOp.SUM, group=group)


def all_reduce_max(t: torch.Tensor, group) -> None:
    assert not t.is_complex()
    dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)


def reduce_sum(t: torch.Tensor, dst_global_rank: int, group) -> None:
    dist.reduce(_commview(t), dst=dst_global_rank, op=dist.ReduceOp.SUM, group=group)


def send(t: torch.Tensor, dst_global_rank: int, group=None, tag: int = 0) -> None:
    dist.send(_commview(t), dst=dst_global_rank, group=group, tag=tag)


def recv(t: torch.Tensor, src_global_rank: int, group=None, tag: int = 0) -> None:
    dist.recv(_commview(t), src=src_global_rank, group=group, tag=tag)
