#!/usr/bin/env python3
"""Communication benchmark (reference ``miniapp/miniapp_communication.cpp``):
times broadcast / all-reduce / reduce / p2p of tile-sized payloads on the
row/col/full communicators (RCCL over xGMI on GPU, gloo on CPU)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from _harness import common_parser, MiniappCtx
from dlaf_amd.comm import collectives as coll


def main():
    opts = common_parser("miniapp_communication").parse_args()
    ctx = MiniappCtx(opts)
    nb = opts.block_size
    t = torch.randn(nb, nb, dtype=torch.float64, device=ctx.device)
    reps = max(1, opts.nruns) * 10
    results = []
    for name, group in [("full", ctx.grid.full_group),
                        ("row", ctx.grid.row_group),
                        ("col", ctx.grid.col_group)]:
        if group is None and name != "full":
            continue
        if not ctx.grid.distributed:
            continue
        for coll_name, fn in [
            ("bcast", lambda: coll.broadcast(t, 0, group)),
            ("allreduce", lambda: coll.all_reduce_sum(t, group)),
        ]:
            for _ in range(3):
                fn()
            ctx.barrier_sync()
            t0 = time.perf_counter()
            for _ in range(reps):
                fn()
            ctx.barrier_sync()
            el = (time.perf_counter() - t0) / reps
            gbps = t.numel() * t.element_size() / el / 1e9
            results.append((name, coll_name, el, gbps))
    if ctx.rank == 0:
        for name, cn, el, gbps in results:
            print(f"[{name}:{cn}] {el*1e6:.1f}us {gbps:.2f}GB/s payload {nb}x{nb} f64")
        if not results:
            print("single rank: no communication to benchmark")
    ctx.finalize()


if __name__ == "__main__":
    main()
