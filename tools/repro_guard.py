import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from dlaf_amd.ops import tile_ops as ops
from dlaf_amd.types import Op

torch.manual_seed(5)
nb, ntiles = 256, 3
dtype = torch.float64
L = torch.tril(torch.randn(nb, nb, dtype=dtype, device="cuda")) + 2*nb*torch.eye(nb, dtype=dtype, device="cuda")
panel0 = torch.randn(ntiles, nb, nb, dtype=dtype, device="cuda")
dinv = ops.dinv_workspace(nb, dtype, "cuda")
bsz = dinv.shape[-1]
ext = ops.get_ext()
for d in range((nb + bsz - 1)//bsz):
    c0 = d*bsz; bs = min(bsz, nb-c0)
    ext.trtri_lower(L[c0:, c0:], dinv[d], bs, L.stride(0), bsz, False)
offs = [i*nb*nb for i in range(ntiles)]
for trial in range(4):
    panel = panel0.clone()
    ops.trsm_panel_right_lowerH(panel, offs, L, dinv, nb, nb, nb)
    torch.cuda.synchronize()
    bad = ~torch.isfinite(panel)
    print(f"trial {trial}: nan count={int(bad.sum())}", flush=True)
    if bad.any():
        idx = bad.nonzero()[:5]
        print("  first bad:", idx.tolist())
# standalone aliased gemm: X = X @ dinv^T, M=256 N=64 K=64 inplace
X0 = torch.randn(ntiles, nb, nb, dtype=dtype, device="cuda")
for trial in range(4):
    X = X0.clone()
    descs = ops.make_descs(offs, offs, [0]*ntiles)
    ops.gemm_fused(X, X, dinv[0], descs, nb, 64, 64, nb, bsz, nb, Op.NoTrans, Op.Trans, 1.0, 0.0, inplace=True)
    torch.cuda.synchronize()
    print(f"alias gemm trial {trial}: nan={int((~torch.isfinite(X)).sum())}", flush=True)
# standalone non-aliased: trsm_upd shape: C=panel+c0, A=panel, B=L, K=c0
Y0 = torch.randn(ntiles, nb, nb, dtype=dtype, device="cuda")
for c0 in [64, 128, 192]:
    Y = Y0.clone()
    descs = ops.make_descs([o + c0 for o in offs], offs, [c0*nb]*ntiles)
    ops.gemm_fused(Y, Y, L, descs, nb, 64, c0, nb, nb, nb, Op.NoTrans, Op.Trans, -1.0, 1.0)
    torch.cuda.synchronize()
    print(f"upd c0={c0}: nan={int((~torch.isfinite(Y)).sum())}", flush=True)
