"""Per-tile RW-chained task DAG on HIP streams — the sender/receiver layer.

Counterpart of the reference's async API: ``Matrix::read()/readwrite()``
senders guarded by ``pika::async_rw_mutex`` (``matrix/tile.h:150-192``) and
``internal::transform`` dispatching to a stream from the pool
(``sender/transform.h:52-110``). MI355X-native shape:

* every resource key (a tile coordinate, a workspace name) carries an RW
  chain: concurrent readers after the last writer; the next writer after
  all readers — exactly the async_rw_mutex ordering;
* ``submit(fn, reads, writes)`` places ``fn(stream)`` on a round-robin HIP
  stream, makes it wait (via events) for the chain heads it depends on, and
  records its completion event into the chains it touches;
* on CPU the same code degrades to immediate in-order execution (events are
  no-ops) so distributed CPU tests exercise identical task bodies.

The production algorithms use phase-fused kernels (one launch per phase —
measured faster than per-tile tasks on MI355X, docs/DESIGN.md §1); this
engine provides the reference's composable async surface for writing NEW
algorithms tile-by-tile, and is exercised by a per-tile Cholesky
(``cholesky_dag``) validated against the fused implementation.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional

import torch

from .streams import get_runtime


class _Chain:
    __slots__ = ("writer", "readers")

    def __init__(self):
        self.writer = None          # event of the last writer
        self.readers: List = []     # events of readers since that writer


class TileDag:
    """RW-chained task submission over a stream pool (one device)."""

    def __init__(self, device: torch.device, n_streams: int = 4):
        self.device = device
        self.gpu = device.type == "cuda"
        self._chains: Dict = {}
        if self.gpu:
            rt = get_runtime(device)
            self._streams = list(rt.np_streams[:max(1, n_streams)])
            cur = torch.cuda.current_stream(device)
            for s in self._streams:
                s.wait_stream(cur)
        else:
            self._streams = [None]
        self._rr = 0

    def _chain(self, key) -> _Chain:
        c = self._chains.get(key)
        if c is None:
            c = _Chain()
            self._chains[key] = c
        return c

    def submit(self, fn, reads: Iterable = (), writes: Iterable = (),
               priority: bool = False) -> None:
        """Run ``fn(stream)`` after its dependencies; order successors."""
        reads = list(reads)
        writes = list(writes)
        if not self.gpu:
            fn(None)
            return
        rt = get_runtime(self.device)
        stream = (rt.hp_streams[0] if priority
                  else self._streams[self._rr % len(self._streams)])
        self._rr += 1
        with torch.cuda.stream(stream):
            for key in reads:
                c = self._chain(key)
                if c.writer is not None:
                    stream.wait_event(c.writer)
            for key in writes:
                c = self._chain(key)
                if c.writer is not None:
                    stream.wait_event(c.writer)
                for ev in c.readers:
                    stream.wait_event(ev)
            fn(stream)
            ev = torch.cuda.Event()
            ev.record(stream)
        for key in reads:
            self._chain(key).readers.append(ev)
        for key in writes:
            c = self._chain(key)
            c.writer = ev
            c.readers = []

    def wait_all(self) -> None:
        """Join every chain back into the caller's stream (the analog of
        ``Matrix::wait_local_tiles``)."""
        if not self.gpu:
            return
        cur = torch.cuda.current_stream(self.device)
        for s in self._streams:
            cur.wait_stream(s)
        rt = get_runtime(self.device)
        cur.wait_stream(rt.hp_streams[0])


def cholesky_dag(mat) -> None:
    """Per-tile right-looking Cholesky over the DAG engine (the reference's
    task shape, ``factorization/cholesky/impl.h:151-189``): one task per
    tile op, ordered purely by read/readwrite chains. Validation target for
    the engine — the production path is the fused ``_cholesky_local``."""
    from ..ops import tile_ops as ops
    from ..types import Op, is_complex

    d = mat.dist
    nt = d.nr_tiles[0]
    opc = Op.ConjTrans if is_complex(mat.dtype) else Op.Trans
    dag = TileDag(mat.device)
    gpu = mat.device.type == "cuda"

    def potrf(k):
        def body(stream):
            ops.potrf_tile(mat.tile((k, k)), None)
        return body

    def trsm(i, k):
        def body(stream):
            diag = mat.tile((k, k))
            t = mat.tile((i, k))
            # A[i,k] <- A[i,k] L_kk^{-H} (right-solve with the upper L^H)
            t.copy_(torch.linalg.solve_triangular(
                torch.tril(diag).mH, t, upper=True, left=False))
        return body

    def gemm(i, j, k):
        def body(stream):
            ops.gemm_tile(mat.tile((i, j)), mat.tile((i, k)),
                          mat.tile((j, k)), Op.NoTrans, opc, -1.0, 1.0)
        return body

    for k in range(nt):
        dag.submit(potrf(k), writes=[(k, k)], priority=True)
        for i in range(k + 1, nt):
            dag.submit(trsm(i, k), reads=[(k, k)], writes=[(i, k)],
                       priority=(i == k + 1))
        for j in range(k + 1, nt):
            for i in range(j, nt):
                dag.submit(gemm(i, j, k), reads=[(i, k), (j, k)],
                           writes=[(i, j)])
    dag.wait_all()
