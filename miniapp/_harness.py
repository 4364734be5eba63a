"""Shared miniapp harness.

Replicates the reference's miniapp contract (``miniapp/include/dlaf/miniapp/
options.h:241-311``, ``miniapp_cholesky.cpp:106-200``): common CLI options,
warmup+run loop with barrier-bracketed timing, human-readable line

    [run] <t>s <gflops>GFlop/s <type><uplo> (m, n) (mb, nb) (r, c) <threads> <backend>

and the machine-readable ``CSVData-2`` row. Multi-rank execution comes from
``python -m torch.distributed.run --nproc-per-node R*C miniapp/miniapp_x.py
--grid-rows R --grid-cols C`` (one rank per GPU over RCCL, gloo on CPU).
"""

from __future__ import annotations

import argparse
import os
import time
from typing import Callable, Optional

import torch
import torch.distributed as dist

from dlaf_amd import CommGrid, Matrix

_DTYPES = {"s": torch.float32, "d": torch.float64,
           "c": torch.complex64, "z": torch.complex128}


def common_parser(name: str, extra: Callable[[argparse.ArgumentParser], None] = None
                  ) -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog=name)
    p.add_argument("--matrix-size", "-m", type=int, default=4096)
    p.add_argument("--block-size", "-b", type=int, default=256)
    p.add_argument("--grid-rows", type=int, default=1)
    p.add_argument("--grid-cols", type=int, default=1)
    p.add_argument("--nruns", type=int, default=1)
    p.add_argument("--nwarmups", type=int, default=1)
    p.add_argument("--check-result", choices=["none", "last", "all"], default="none")
    p.add_argument("--type", choices=list(_DTYPES), default="d")
    p.add_argument("--backend", choices=["default", "mc", "gpu"], default="default")
    p.add_argument("--local", action="store_true")
    p.add_argument("--csv-output", action="store_true")
    if extra:
        extra(p)
    return p


class MiniappCtx:
    def __init__(self, opts):
        self.opts = opts
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.rank = int(os.environ.get("RANK", "0"))
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        if opts.backend == "mc":
            self.device = torch.device("cpu")
        elif opts.backend == "gpu":
            assert torch.cuda.is_available(), "--backend gpu needs a GPU"
            self.device = torch.device(f"cuda:{local_rank}")
        else:
            self.device = (torch.device(f"cuda:{local_rank}")
                           if torch.cuda.is_available() else torch.device("cpu"))
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        if self.world_size > 1 and not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            backend = "nccl" if self.device.type == "cuda" else "gloo"
            dist.init_process_group(backend, rank=self.rank, world_size=self.world_size)
        assert opts.grid_rows * opts.grid_cols == self.world_size, \
            f"grid {opts.grid_rows}x{opts.grid_cols} != world {self.world_size}"
        self.grid = CommGrid(opts.grid_rows, opts.grid_cols, device=self.device)
        self.dtype = _DTYPES[opts.type]

    @property
    def comm_grid(self) -> Optional[CommGrid]:
        return None if self.opts.local else self.grid

    def barrier_sync(self):
        if self.grid.distributed:
            dist.barrier()
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def finalize(self):
        if self.grid.distributed:
            dist.destroy_process_group()


def run_miniapp(name: str, setup: Callable, run: Callable, flops: Callable,
                check: Optional[Callable] = None,
                extra: Callable[[argparse.ArgumentParser], None] = None,
                extra_fields: Callable = None) -> None:
    """setup(ctx) -> state; run(ctx, state) -> result; flops(ctx) -> float or None;
    check(ctx, state, result) -> residual float (printed)."""
    opts = common_parser(name, extra).parse_args()
    ctx = MiniappCtx(opts)
    backend_name = "GPU" if ctx.device.type == "cuda" else "MC"
    for run_index in range(-opts.nwarmups, opts.nruns):
        state = setup(ctx)
        ctx.barrier_sync()
        t0 = time.perf_counter()
        result = run(ctx, state)
        ctx.barrier_sync()
        elapsed = time.perf_counter() - t0
        f = flops(ctx)
        gflops = (f / elapsed / 1e9) if f else float("nan")
        if ctx.rank == 0 and run_index >= 0:
            o = opts
            print(f"[{run_index}] {elapsed}s {gflops}GFlop/s {o.type}L "
                  f"({o.matrix_size}, {o.matrix_size}) ({o.block_size}, {o.block_size}) "
                  f"({o.grid_rows}, {o.grid_cols}) {torch.get_num_threads()} {backend_name}",
                  flush=True)
            if o.csv_output:
                row = (f"CSVData-2, run, {run_index}, time, {elapsed}, GFlops, {gflops}, "
                       f"type, {o.type}, UpLo, L, matrixsize, {o.matrix_size}, "
                       f"blocksize, {o.block_size}, comm_rows, {o.grid_rows}, "
                       f"comm_cols, {o.grid_cols}, threads, {torch.get_num_threads()}, "
                       f"backend, {backend_name}")
                if extra_fields:
                    row += ", " + extra_fields(ctx)
                print(row, flush=True)
        do_check = (opts.check_result == "all" or
                    (opts.check_result == "last" and run_index == opts.nruns - 1))
        if do_check and check is not None:
            resid = check(ctx, state, result)
            if ctx.rank == 0:
                print(f"[{run_index}] check residual = {resid:.3e}", flush=True)
    ctx.finalize()


def random_spd(ctx, n=None, nb=None) -> Matrix:
    from dlaf_amd.matrix import util as mutil
    n = n or ctx.opts.matrix_size
    nb = nb or ctx.opts.block_size
    m = Matrix.create(n, n, nb, nb, dtype=ctx.dtype, device=ctx.device, grid=ctx.grid)
    mutil.set_random_hermitian_positive_definite(m, seed=0)
    return m


def random_herm(ctx, n=None, nb=None) -> Matrix:
    from dlaf_amd.matrix import util as mutil
    n = n or ctx.opts.matrix_size
    nb = nb or ctx.opts.block_size
    m = Matrix.create(n, n, nb, nb, dtype=ctx.dtype, device=ctx.device, grid=ctx.grid)
    mutil.set_random_hermitian(m, seed=0)
    return m


def random_general(ctx, m_, n_, nb=None) -> Matrix:
    from dlaf_amd.matrix import util as mutil
    nb = nb or ctx.opts.block_size
    m = Matrix.create(m_, n_, nb, nb, dtype=ctx.dtype, device=ctx.device, grid=ctx.grid)
    mutil.set_random(m, seed=1)
    return m
