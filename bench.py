#!/usr/bin/env python3
"""Flagship benchmark: fp64 tiled Cholesky (POTRF), N=32768, nb=512.

BASELINE.json config 2 ("fp64 Cholesky (POTRF) N=32768 nb=512 on 1 MI355X") and
its multi-GPU strong-scaling variants (grid 1x2 / 2x2 / 2x4 over RCCL/xGMI).
Metric: GFlop/s with the reference's flop count n^3/3 (n^3/6 mul + n^3/6 add,
``miniapp/miniapp_cholesky.cpp:157-163``; complex weights would be 2/6).

One step = restore the matrix from a device-resident pristine copy (D2D,
~1-2%% of step time, included in the timing) + full in-place factorization.
Protocol: W untimed warmup steps, then exactly K timed steps bracketed by a
barrier + torch.cuda.synchronize on both sides; elapsed time is the MAX over
ranks; rank 0 prints one JSON line.
"""

import argparse
import json
import os
import time

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--n", type=int, default=32768)
    p.add_argument("--nb", type=int, default=512)
    p.add_argument("--device", type=str, default=None, help="debug override (cpu)")
    args = p.parse_args()

    import torch.distributed as dist

    from dlaf_amd import Matrix, CommGrid, UpLo, cholesky_factorization
    from dlaf_amd.matrix import util as mutil

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    if args.device == "cpu":
        device = torch.device("cpu")
    else:
        assert torch.cuda.is_available(), "bench needs a GPU (or --device cpu)"
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)

    if world_size > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = "nccl" if device.type == "cuda" else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world_size)

    grids = {1: (1, 1), 2: (1, 2), 4: (2, 2), 8: (2, 4)}
    gr, gc = grids.get(world_size, (1, world_size))
    grid = CommGrid(gr, gc, device=device)

    n, nb = args.n, args.nb
    mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device=device, grid=grid)
    mutil.set_random_hermitian_positive_definite(mat, seed=42)
    pristine = mat.storage.clone()

    def barrier_sync():
        if world_size > 1:
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    def step():
        mat.storage.copy_(pristine)
        cholesky_factorization(UpLo.Lower, mat, grid)

    for _ in range(args.warmup):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world_size > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    flops = n ** 3 / 3.0  # fp64 POTRF: n^3/6 mul + n^3/6 add
    gflops = flops / (elapsed / args.steps) / 1e9

    if rank == 0:
        print(json.dumps({
            "metric": "cholesky_fp64_gflops",
            "value": gflops,
            "unit": "GFlop/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic random SPD (diagonally dominant), restore included in step",
            "config": {
                "model": "fp64 Cholesky (POTRF)",
                "n": n,
                "nb": nb,
                "parallelism": f"grid{gr}x{gc}",
                "global_batch": 1,
                "seq_len": n,
            },
        }))

    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
