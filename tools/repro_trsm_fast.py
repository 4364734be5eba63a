"""Localize TRSM LLN fast-path failures: run all dtype/diag combos at
non-divisible sizes, compare against the CPU reference, report per-tile-row
max error to find the first bad row."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dlaf_amd.types import Side, UpLo, Op, Diag
from dlaf_amd.matrix.matrix import Matrix
from dlaf_amd.matrix import util as mutil
from dlaf_amd.algs.triangular import triangular_solver

m, n, nb = 1100, 900, 256
for dtype in (torch.float64, torch.complex128):
    for diag in (Diag.NonUnit, Diag.Unit):
        A = Matrix.create(m, m, nb, nb, dtype=dtype, device="cuda")
        B = Matrix.create(m, n, nb, nb, dtype=dtype, device="cuda")
        mutil.set_random_hermitian_positive_definite(A, seed=5)
        mutil.set_random(B, seed=6)
        a, b0 = A.to_global().cpu(), B.to_global().cpu()
        triangular_solver(Side.Left, UpLo.Lower, Op.NoTrans, diag, 0.5, A, B)
        torch.cuda.synchronize()
        tri = torch.tril(a)
        if diag == Diag.Unit:
            tri = tri - torch.diag_embed(tri.diagonal()) + torch.eye(m, dtype=tri.dtype)
        want = torch.linalg.solve(tri, 0.5 * b0)
        got = B.to_global().cpu()
        err = (got - want).abs().max().item()
        rows = [float((got[i*nb:(i+1)*nb] - want[i*nb:(i+1)*nb]).abs().max())
                for i in range((m + nb - 1)//nb)]
        print(f"{dtype} {diag}: err={err:.3e} per-row={['%.1e'%r for r in rows]}",
              flush=True)
