"""Spawn-based multi-process harness for distributed CPU tests (gloo).

The analog of the reference's 6-rank MPI test fixture
(``test/include/dlaf_test/comm_grids/grids_6_ranks.h``): distributed algorithm
tests run here, on CPU, with real collectives — the GPU path shares the same
comm code with the RCCL backend.
"""

from __future__ import annotations

import os
import pickle
import tempfile

import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, port, fn, args, result_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        out = fn(rank, world_size, *args)
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(out, f)
    finally:
        # drain all in-flight gloo work before teardown: destroying the
        # process group while a peer still communicates aborts in the gloo
        # device thread ("terminate called without an active exception")
        try:
            dist.barrier()
        except Exception:
            pass
        dist.destroy_process_group()


def run_distributed(fn, world_size, args=()):
    """Run ``fn(rank, world_size, *args)`` in ``world_size`` processes.

    Returns the list of per-rank return values (must be picklable).
    Failures in any rank propagate as ProcessRaisedException.
    """
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    with tempfile.TemporaryDirectory() as result_dir:
        mp.spawn(
            _worker,
            args=(world_size, port, fn, args, result_dir),
            nprocs=world_size,
            join=True,
        )
        results = []
        for r in range(world_size):
            with open(os.path.join(result_dir, f"rank{r}.pkl"), "rb") as f:
                results.append(pickle.load(f))
        return results
