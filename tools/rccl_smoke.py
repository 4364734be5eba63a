"""2-rank RCCL smoke on a single GPU (both ranks -> cuda:0): exercises the
real NCCL/RCCL code path of the distributed algorithms (groups, broadcasts,
stream-ordered collectives) that the multi-GPU SCALE run uses."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
import torch.distributed as dist

rank = int(os.environ["RANK"]); ws = int(os.environ["WORLD_SIZE"])
torch.cuda.set_device(0)
dist.init_process_group("nccl", rank=rank, world_size=ws)

from dlaf_amd import Matrix, CommGrid, UpLo, cholesky_factorization, triangular_solver, Side, Op, Diag
from dlaf_amd.matrix import util as mutil

grid = CommGrid(1, ws, device=torch.device("cuda", 0))
n, nb = 4096, 512
mat = Matrix.create(n, n, nb, nb, dtype=torch.float64, device="cuda:0", grid=grid)
mutil.set_random_hermitian_positive_definite(mat, seed=42)
a = mat.to_global()
cholesky_factorization(UpLo.Lower, mat, grid)
torch.cuda.synchronize()
L = torch.tril(mat.to_global())
res = (a - L @ L.mH).abs().max().item() / a.abs().max().item()
print(f"rank {rank}: chol residual {res:.3e}", flush=True)
assert res < 1e-12, res

B = Matrix.create(n, 2048, nb, nb, dtype=torch.float64, device="cuda:0", grid=grid)
mutil.set_random(B, seed=1)
b0 = B.to_global()
triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, Diag.NonUnit, 1.0, mat, B, grid)
torch.cuda.synchronize()
got = torch.tril(L) @ B.to_global()
err = (got - b0).abs().max().item() / b0.abs().max().item()
print(f"rank {rank}: trsm residual {err:.3e}", flush=True)
assert err < 1e-10, err
dist.destroy_process_group()
print(f"rank {rank}: RCCL smoke OK", flush=True)
