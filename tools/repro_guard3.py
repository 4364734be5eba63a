"""Replicate potrf_tile(512)'s internal gemm sequence step by step, saving
intermediates, to localize the guard-free divergence.

Usage: DLAF_GEMM_FULLOPT={0|1} python tools/repro_guard3.py /tmp/out_{0|1}.pt
Then:  python tools/repro_guard3.py --compare /tmp/out_0.pt /tmp/out_1.pt
"""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch

if sys.argv[1] == "--compare":
    a = torch.load(sys.argv[2], weights_only=True)
    b = torch.load(sys.argv[3], weights_only=True)
    for k in a:
        d = (a[k] - b[k]).abs().max().item()
        na = int((~torch.isfinite(a[k])).sum())
        nb_ = int((~torch.isfinite(b[k])).sum())
        print(f"{k}: maxdiff={d:.3e} nonfinite: {na} vs {nb_}")
    sys.exit(0)

from dlaf_amd.ops import tile_ops as ops
from dlaf_amd.types import Op

out = {}
torch.manual_seed(3)
n, ld = 512, 512
dtype = torch.float64
a = torch.randn(n, n, dtype=dtype)
a = a @ a.mT + n * torch.eye(n, dtype=dtype)
A = a.cuda()
bsz = 64
dinv = ops.dinv_workspace(n, dtype, "cuda")
ext = ops.get_ext()
for d in range((n + bsz - 1) // bsz):
    c0 = d * bsz
    bs = min(bsz, n - c0)
    ext.factor_invert_block(A[c0:, c0:], bs, A.stride(0), dinv[d], True)
    torch.cuda.synchronize()
    out[f"factor{d}"] = A.cpu().clone()
    rows_below = n - c0 - bs
    if rows_below <= 0:
        continue
    panel_off = (c0 + bs) * ld + c0
    trail_off = (c0 + bs) * ld + (c0 + bs)
    # panel: X = A21 @ dinv^T (in place)
    ops.gemm_fused(A, A, dinv[d], ops.make_descs([panel_off], [panel_off], [0]),
                   rows_below, bs, bs, ld, bsz, ld, Op.NoTrans, Op.Trans,
                   1.0, 0.0, inplace=True)
    torch.cuda.synchronize()
    out[f"panel{d}"] = A.cpu().clone()
    # trailing: A22 -= X X^T
    ops.gemm_fused(A, A, A, ops.make_descs([trail_off], [panel_off], [panel_off]),
                   rows_below, rows_below, bs, ld, ld, ld, Op.NoTrans, Op.Trans,
                   -1.0, 1.0)
    torch.cuda.synchronize()
    out[f"trail{d}"] = A.cpu().clone()
torch.save(out, sys.argv[1])
ref = torch.linalg.cholesky(a)
err = (torch.tril(A.cpu()) - ref).abs().max().item()
print("final err:", err)
