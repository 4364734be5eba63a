"""Minimal-config hunt for the guard-free divergence (runs both modes are
compared against CPU reference; FULLOPT must be 1 in env)."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from dlaf_amd.ops import tile_ops as ops
from dlaf_amd.types import Op

def one(M, K, lda, off, inplace, tag):
    torch.manual_seed(7)
    total = off + M * lda + 64  # enough storage
    base = torch.randn(total, dtype=torch.float64, device="cuda")
    B = torch.randn(64, K, dtype=torch.float64, device="cuda")  # dinv-like (ldb=K? use bsz)
    Bm = torch.randn(64, 64, dtype=torch.float64, device="cuda")
    A_cpu = base.cpu().clone()
    descs = ops.make_descs([off], [off], [0])
    ops.gemm_fused(base, base, Bm, descs, M, 64, K, lda, 64, lda,
                   Op.NoTrans, Op.Trans, 1.0, 0.0, inplace=inplace)
    torch.cuda.synchronize()
    # reference: rows i of sub-block [off + i*lda + 0..K) @ Bm[0:64,0:K]^T
    sub = torch.stack([A_cpu[off + i*lda: off + i*lda + K] for i in range(M)])
    ref = sub @ Bm.cpu()[:, :K].mT
    got = torch.stack([base.cpu()[off + i*lda: off + i*lda + 64] for i in range(M)])
    err = (got - ref).abs().max().item()
    print(f"{tag}: M={M} K={K} lda={lda} off={off} inplace={inplace} err={err:.3e}", flush=True)

for M in [128, 256, 384]:
    for lda in [512, 384, 64]:
        if lda < 64: continue
        for off in [0, 65600]:
            for inpl in [True, False]:
                one(M, 64, lda, off, inpl, "cfg")
