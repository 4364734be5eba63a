"""Reproduce the guard-free NaN: run the op-combo gemms first (as pytest
does), then the trsm panel — in one process."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from dlaf_amd.ops import tile_ops as ops
from dlaf_amd.types import Op

def ops_phase():
    for dtype in [torch.float64, torch.float32]:
        for opA in [Op.NoTrans, Op.Trans, Op.ConjTrans]:
            for opB in [Op.NoTrans, Op.Trans, Op.ConjTrans]:
                torch.manual_seed(0)
                M, N, K = 256, 128, 192
                a_shape = (M, K) if opA is Op.NoTrans else (K, M)
                b_shape = (K, N) if opB is Op.NoTrans else (N, K)
                A = torch.randn(a_shape, dtype=dtype, device="cuda")
                B = torch.randn(b_shape, dtype=dtype, device="cuda")
                C = torch.randn((M, N), dtype=dtype, device="cuda")
                descs = ops.make_descs([0], [0], [0])
                ops.gemm_fused(C, A, B, descs, M, N, K, a_shape[1], b_shape[1], N, opA, opB, 1.5, 0.25)
    torch.cuda.synchronize()

def trsm_phase(tag):
    torch.manual_seed(5)
    for dtype in [torch.float64, torch.float32]:
        nb, ntiles = 256, 3
        L = torch.tril(torch.randn(nb, nb, dtype=dtype, device="cuda")) + 2*nb*torch.eye(nb, dtype=dtype, device="cuda")
        panel = torch.randn(ntiles, nb, nb, dtype=dtype, device="cuda")
        dinv = ops.dinv_workspace(nb, dtype, "cuda")
        bsz = dinv.shape[-1]
        ext = ops.get_ext()
        for d in range((nb + bsz - 1)//bsz):
            c0 = d*bsz; bs = min(bsz, nb-c0)
            ext.trtri_lower(L[c0:, c0:], dinv[d], bs, L.stride(0), bsz, False)
        offs = [i*nb*nb for i in range(ntiles)]
        ops.trsm_panel_right_lowerH(panel, offs, L, dinv, nb, nb, nb)
        torch.cuda.synchronize()
        Lh = L.cpu().double().mT if not dtype.is_complex else L.cpu().mH
        bad = int((~torch.isfinite(panel)).sum())
        ref = torch.linalg.solve_triangular(Lh, panel.cpu().double()[0]*0 + 1, upper=True, left=False)  # dummy
        print(f"[{tag}] {dtype}: nan={bad}", flush=True)

def potrf_phase(tag):
    for dtype in [torch.float64, torch.float32]:
        torch.manual_seed(3)
        n = 512
        a = torch.randn(n, n, dtype=dtype).cpu()
        a = a @ a.mT + n*torch.eye(n, dtype=dtype)
        tile = a.cuda()
        ops.potrf_tile(tile)
        torch.cuda.synchronize()
        ref = torch.linalg.cholesky(a.double())
        err = (torch.tril(tile.cpu().double()) - ref).abs().max().item()
        print(f"[{tag}] potrf512 {dtype}: err={err:.3e}", flush=True)

for r in range(3):
    ops_phase()
    trsm_phase(f"round{r}")
    potrf_phase(f"round{r}")
