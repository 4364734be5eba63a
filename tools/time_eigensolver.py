"""Timing breakdown of the eigensolver pipeline on GPU (SYEV config 4 shape)."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from dlaf_amd import Matrix, UpLo
from dlaf_amd.matrix import util as mutil
from dlaf_amd.algs.red2band import reduction_to_band, bt_reduction_to_band
from dlaf_amd.algs.band2tridiag import band_to_tridiagonal, bt_band_to_tridiagonal
from dlaf_amd.algs.tridiag_dc import tridiagonal_eigensolver
from dlaf_amd.algs.eigensolver import get_band_size

n = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
nb = int(sys.argv[2]) if len(sys.argv) > 2 else 512
band = int(sys.argv[3]) if len(sys.argv) > 3 else get_band_size(nb)
dtype = {"d": torch.float64, "z": torch.complex128,
         "s": torch.float32, "c": torch.complex64}[
             sys.argv[4] if len(sys.argv) > 4 else "d"]
dev = "cuda" if torch.cuda.is_available() else "cpu"
mat = Matrix.create(n, n, nb, nb, dtype=dtype, device=dev)
mutil.set_random_hermitian(mat, seed=1)
a0 = None
if n <= 4096:
    a0 = mat.to_global()
    a0 = torch.tril(a0) + torch.tril(a0, -1).mH

def sync():
    if dev == "cuda":
        torch.cuda.synchronize()

stamps = [("start", time.perf_counter())]
refl = reduction_to_band(mat, band); sync(); stamps.append(("red2band", time.perf_counter()))
tri = band_to_tridiagonal(UpLo.Lower, band, mat); sync(); stamps.append(("band2tridiag", time.perf_counter()))
w, E_real = tridiagonal_eigensolver(tri.d, tri.e, device=mat.device); sync(); stamps.append(("tridiag_dc", time.perf_counter()))
E = E_real.to(mat.dtype).contiguous()
if tri.phases is not None:
    pass  # phases applied inside bt
bt_band_to_tridiagonal(E, tri); sync(); stamps.append(("bt_band2tridiag", time.perf_counter()))
bt_reduction_to_band(E, mat, refl); sync(); stamps.append(("bt_red2band", time.perf_counter()))
tot = stamps[-1][1] - stamps[0][1]
print(f"HEEV[{mat.dtype}] n={n} nb={nb} band={band} dev={dev}: total {tot:.2f}s")
for (nm, t1), (_, t0) in zip(stamps[1:], stamps[:-1]):
    print(f"  {nm:16s} {t1-t0:8.2f}s")
if a0 is not None:
    res = (a0 @ E - E @ torch.diag(w.to(E.dtype))).abs().max().item()
    orth = (E.mH @ E - torch.eye(n, dtype=E.dtype, device=E.device)).abs().max().item()
    print(f"  res={res:.2e} orth={orth:.2e}")
