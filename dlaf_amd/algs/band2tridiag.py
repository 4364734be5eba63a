"""Band -> tridiagonal reduction (bulge chasing) and its back-transform.

Counterpart of ``eigensolver/band_to_tridiag/mc.h`` (local 663-989; CPU-only
compute, as the reference: the band is copied D2H, chased on the host, and
the reflectors are applied to the eigenvectors on the GPU afterwards) and of
``eigensolver/bt_band_to_tridiag/impl.h``.

The chase itself is the native C++ engine (``csrc/band_chase.cpp``). The
back-transform exploits that the reflectors of ONE sweep act on DISJOINT
row blocks of length ``band``: each sweep becomes two batched GEMV-shaped
device ops over an [nblocks, band, nE] view of E (sweeps applied in reverse
chronological order).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import os

import torch

from ..types import UpLo
from ..matrix.matrix import Matrix
from ..ops._ext import get_ext


@dataclass
class TridiagResult:
    """d, e: real tridiagonal; vstore/offsets: compact chase reflectors."""
    d: torch.Tensor            # [n] real
    e: torch.Tensor            # [n-1] real
    band: int
    n: int
    vstore: torch.Tensor       # [slots, band+1]: tau, v0=1, v-tail (CPU)
    offsets: torch.Tensor      # [n] int64 slot offset per sweep
    nslots: torch.Tensor       # [n] int64 slots per sweep
    phases: Optional[torch.Tensor] = None   # [n] unitary diag (complex input)


def _slot_counts(n: int, b: int) -> torch.Tensor:
    """Reflector slots per sweep: 1 + number of chase steps with m > 1
    (closed form of the do_step loop; sweep s chases while
    1 + s + t*b <= n - b - 2)."""
    counts = torch.zeros(n, dtype=torch.int64)
    if n > 2:
        s = torch.arange(n - 2)
        tmax = torch.div(n - b - 3 - s, b, rounding_mode="floor")
        counts[: n - 2] = 1 + torch.clamp(tmax + 1, min=0)
    return counts



def chase_band(store: torch.Tensor, b: int) -> TridiagResult:
    """Run the bulge chase on a compact band ``store`` [n, 2b] (consumed in
    place) and package the TridiagResult.

    GPU path (``csrc/chase_gpu.hip``, the DEFAULT on device since round 2;
    ``DLAF_GPU_CHASE=1/0`` forces): persistent wavefront workgroups, one
    UNIT on 256 threads (4 waves splitting the q/p loops with LDS partial
    reductions — the round-1 single-wave form left ~1 wave per CU busy),
    with the band, reflectors and flags resident in HBM; sc1 write-through
    window publish. The reference keeps this stage on the host
    (``band_to_tridiag/mc.h:666-693``). Deterministic (fixed reduction
    order), so rank-replicated distributed use stays lockstep-identical;
    bounded spins surface scheduling pathologies as a CPU-fallback warning,
    never a hang. Measured n=20000 f64: 3.6 s vs 4.1 s CPU wavefront;
    n=16384 c128: 2.9 s vs 10.8 s.
    """
    from ..ops._ext import get_ext
    n = store.shape[0]
    is_cplx = store.is_complex()
    counts = _slot_counts(n, b)
    offsets = torch.zeros(n, dtype=torch.int64)
    if n > 1:
        offsets[1:] = torch.cumsum(counts, 0)[:-1]
    total = int(counts.sum().item())
    vstore = None
    env = os.environ.get("DLAF_GPU_CHASE")
    # default: GPU chase for every dtype (round-2 256-thread units +
    # write-through publish: f64 3.6 s vs 4.1 CPU, c128 2.9 vs 10.8 at the
    # BASELINE shapes; docs/DESIGN.md has the progression)
    want_gpu = True if env is None else env == "1"
    if n > 2 and store.is_cuda and b <= 64 and want_gpu:
        dev = store.device
        vstore = torch.zeros((max(total, 1), b + 1), dtype=store.dtype, device=dev)
        done = torch.zeros(n, dtype=torch.int32, device=dev)
        abortf = torch.zeros(1, dtype=torch.int32, device=dev)
        backup = store.clone()
        get_ext().band_chase_gpu(store, b, vstore, offsets.to(dev), done, abortf)
        torch.cuda.synchronize()
        if int(abortf.item()) != 0:
            import warnings
            warnings.warn("GPU band chase aborted (bounded spin exhausted); "
                          "falling back to the CPU chase")
            store.copy_(backup)
            vstore = None
        del backup, done, abortf
    if vstore is None:
        store_h = store.cpu() if store.is_cuda else store
        vstore = torch.zeros((max(total, 1), b + 1), dtype=store_h.dtype)
        if n > 2:
            nthr = int(os.environ.get("DLAF_CHASE_THREADS", "0"))
            get_ext().band_chase(store_h, b, vstore, offsets, nthr)
        store = store_h
    dvec = store[:, 0]
    evec = store[: n - 1, 1] if n > 1 else store[:0, 1]
    phases = None
    if is_cplx:
        e_abs = evec.abs()
        safe = torch.where(e_abs > 0, e_abs, torch.ones_like(e_abs))
        factor = torch.where(e_abs > 0, evec / safe, torch.ones_like(evec))
        ph = torch.ones(n, dtype=store.dtype, device=store.device)
        if n > 1:
            ph[1:] = torch.cumprod(factor, 0)
            ph = ph / ph.abs()          # keep strictly unit modulus
        phases = ph
        d_real = dvec.real.cpu().clone()
        e_real = e_abs.to(d_real.dtype).cpu()
    else:
        d_real = dvec.cpu().clone()
        e_real = evec.cpu().clone()
    return TridiagResult(d=d_real, e=e_real, band=b, n=n, vstore=vstore,
                         offsets=offsets, nslots=counts, phases=phases)


def band_to_tridiagonal(uplo: UpLo, band: int, mat: Matrix) -> TridiagResult:
    """Reduce the band part of ``mat`` (lower, bandwidth ``band``) to real
    tridiagonal form; returns the tridiagonal and the chase reflectors."""
    assert uplo == UpLo.Lower
    d = mat.dist
    n = d.m
    assert d.m == d.n
    g = mat.grid
    assert g is None or not g.distributed, \
        "distributed band_to_tridiagonal lands with the distributed eigensolver"
    A = mat.to_global()
    b = band
    ld = 2 * b
    # extract the compact band on the device, move only n x 2b to the host
    store_dev = torch.zeros((n, ld), dtype=A.dtype, device=A.device)
    for dd in range(min(b, n - 1) + 1):
        store_dev[: n - dd, dd] = torch.diagonal(A, -dd)
    del A
    return chase_band(store_dev, b)


def bt_band_to_tridiagonal(E: torch.Tensor, tri: TridiagResult,
                           group_size: Optional[int] = None) -> None:
    """Back-transform E <- Q (D E) where A_band = Q (D T_real D^H) Q^H.

    E: [n, nE] device tensor of tridiagonal eigenvectors, updated in place.

    Reference: ``eigensolver/bt_band_to_tridiag/impl.h:59-1031`` — including
    its HH-apply GROUPING (``hh_apply_group_size``, tune.h): reflectors of
    ``group_size`` consecutive sweeps at the same chase depth k form one
    staircase V panel applied via a compact-WY T factor (3 GEMMs), so E is
    traversed once per GROUP instead of once per sweep — arithmetic intensity
    up by group_size. Ordering: groups in reverse sweep order; within a
    group, windows k ascending (cross-window overlapping reflector pairs are
    exactly those the ascending order keeps correctly ordered; the disjoint
    pairs commute).
    """
    from contextlib import nullcontext as _nullcontext

    from ..config import get_tune_parameters
    if group_size is None:
        group_size = get_tune_parameters().bt_band_to_tridiag_hh_apply_group_size
    merge = max(1, get_tune_parameters().bt_band_to_tridiag_window_merge)
    n, nE = E.shape
    b = tri.band
    dev = E.device
    if tri.phases is not None:
        E *= tri.phases.to(dev).unsqueeze(1)
    if n <= 2 or tri.vstore.numel() == 0:
        return
    V = tri.vstore.to(dev)
    offsets = tri.offsets.to(dev)
    counts = tri.nslots.to(dev)
    nsweeps = n - 2
    G = max(1, min(group_size, nsweeps))
    H = b + G - 1                       # window height
    Epad = torch.zeros((n + b + G, nE), dtype=E.dtype, device=dev)
    Epad[:n] = E
    eyeG = torch.eye(G, dtype=E.dtype, device=dev)
    group_starts = list(range(0, nsweeps, G))
    # Column-strip parallelism (tunable, default OFF): the window chain only
    # touches rows, so E column strips are independent. Measured at n=20000:
    # strips LOSE (3.6 -> 4.1 s at 2 strips) — the chain is host-launch
    # bound (~70k small GEMMs), and strips double the launch count. Kept as
    # DLAF_BT_STREAMS for wider-nE regimes.
    import os as _os
    n_strips = int(_os.environ.get("DLAF_BT_STREAMS", "1"))
    use_strips = (dev.type == "cuda" and n_strips > 1 and merge == 1
                  and nE >= 2 * n_strips)
    if use_strips:
        from ..runtime.streams import get_runtime
        rt = get_runtime(dev)
        streams = [rt.np_streams[i % len(rt.np_streams)]
                   for i in range(n_strips)]
        cur = torch.cuda.current_stream(dev)
        for st_ in streams:
            st_.wait_stream(cur)
        bounds = [(nE * i) // n_strips for i in range(n_strips + 1)]
    else:
        streams = [None]
        bounds = [0, nE]
    for s0 in reversed(group_starts):
        Gc = min(G, nsweeps - s0)       # sweeps in this group
        counts_g = counts[s0:s0 + Gc]
        nwin = int(counts_g.max())
        if nwin == 0:
            continue
        _prep_ctx = (torch.cuda.stream(streams[0]) if use_strips
                     else _nullcontext())
        with _prep_ctx:
            # staircase V panels for all windows: Vg[k, row, g]. Assembled
            # with ONE gather+scatter per group: the per-sweep slice loop
            # (2 copies + 2 host syncs per sweep) was ~40k tiny device
            # copies per f64 n=20000 solve (profiles/
            # bench_full_r2_kernel_stats.md: 199740 __amd_rocclr_copyBuffer)
            Vg = torch.zeros((nwin, H, G), dtype=E.dtype, device=dev)
            taus_g = torch.zeros((nwin, G), dtype=E.dtype, device=dev)
            cg = counts_g.long()
            gidx = torch.repeat_interleave(torch.arange(Gc, device=dev), cg)
            if gidx.numel():
                starts = torch.cumsum(cg, 0) - cg
                kidx = (torch.arange(gidx.numel(), device=dev)
                        - starts[gidx])
                srow = offsets[s0:s0 + Gc].long()[gidx] + kidx
                taus_g[kidx, gidx] = V[srow, 0]
                # Vg[kidx, gidx + r, gidx] for r in [0, b)
                rr = torch.arange(b, device=dev)
                dst = ((kidx * H + gidx) * G + gidx).unsqueeze(1) \
                    + rr.unsqueeze(0) * G
                Vg.view(-1)[dst.reshape(-1)] = V[srow, 1:].reshape(-1)
            # batched T factors: T = inv(diag(1/tau) + striu(V^H V)); tau=0 rows/cols vanish
            Gram = Vg.mH @ Vg                                   # [nwin, G, G]
            zc = taus_g == 0
            safe = torch.where(zc, torch.ones_like(taus_g), taus_g)
            M = torch.triu(Gram, 1) + torch.diag_embed(1.0 / safe)
            mask = zc.unsqueeze(1) | zc.unsqueeze(2)
            M = torch.where(mask, torch.zeros_like(M), M)
            M = M + torch.diag_embed(torch.where(zc, torch.ones_like(safe),
                                                 torch.zeros_like(safe)))
            T = torch.linalg.solve_triangular(M, eyeG.expand(nwin, G, G).contiguous(), upper=True)
            T = torch.where(mask, torch.zeros_like(T), T)
        # Merge m consecutive windows into one block-WY apply: the product
        # Q_{k+m-1}...Q_k of compact-WY transforms is itself compact-WY with
        #   Vc = [V_k .. V_{k+m-1}] (window j shifted j*b rows down) and
        #   Tc[j, :j] = -T_j (V_j^H [V_k..V_{j-1}]) Tc[:j, :j]
        # (windows with row distance >= ceil(H/b) have zero overlap, so the
        # cross-Grams S vanish beyond two sub-block-diagonals). This keeps
        # the ordering constraint exactly while making the E-traversal GEMMs
        # m-times wider and the serial window chain m-times shorter.
        m = max(1, min(merge, nwin))
        if m > 1:
            Hm = H + (m - 1) * b
            nmerge = (nwin + m - 1) // m
            # pad the window axis to a multiple of m with zero windows
            if nmerge * m != nwin:
                pad = nmerge * m - nwin
                Vg = torch.cat([Vg, torch.zeros((pad, H, G), dtype=Vg.dtype, device=dev)])
                T = torch.cat([T, torch.zeros((pad, G, G), dtype=T.dtype, device=dev)])
            # Vc[km, row, j*G+g] = Vg[km*m+j, row - j*b, g]
            Vgm = Vg.view(nmerge, m, H, G)
            Vc = torch.zeros((nmerge, Hm, m * G), dtype=Vg.dtype, device=dev)
            for j in range(m):
                Vc[:, j * b:j * b + H, j * G:(j + 1) * G] = Vgm[:, j]
            Tm = T.view(nmerge, m, G, G)
            Tc = torch.zeros((nmerge, m * G, m * G), dtype=T.dtype, device=dev)
            for j in range(m):
                Tc[:, j * G:(j + 1) * G, j * G:(j + 1) * G] = Tm[:, j]
            for j in range(1, m):
                # S_ji = V_j^H V_i with row alignment (j-i)*b; zero if no overlap
                S = torch.zeros((nmerge, G, j * G), dtype=T.dtype, device=dev)
                for i in range(j):
                    d0 = (j - i) * b
                    if d0 < H:
                        S[:, :, i * G:(i + 1) * G] = (
                            Vgm[:, j, :H - d0].mH @ Vgm[:, i, d0:])
                row = -(Tm[:, j] @ (S @ Tc[:, :j * G, :j * G]))
                Tc[:, j * G:(j + 1) * G, :j * G] = row
            for km in range(nmerge):
                k0 = km * m
                mc = min(m, nwin - k0)
                base = 1 + s0 + k0 * b
                hseg = H + (mc - 1) * b
                w = mc * G
                seg = Epad[base:base + hseg]
                Vck = Vc[km, :hseg, :w]
                W = Tc[km, :w, :w] @ (Vck.mH @ seg)
                seg -= Vck @ W
        else:
            # whole-group single-launch kernel (csrc/bt_apply.hip): each
            # workgroup owns a 16-column slice and marches the full window
            # chain in LDS — replaces ~3*nwin GEMM launches per group
            # (launch-bound, measured) with ONE.
            # f64: always (the kernel splits partial CU passes into a CW=32
            # tail launch). c128: 512-thread round-2 kernel.
            if (dev.type == "cuda"
                    and E.dtype in (torch.float64, torch.complex128)
                    and G % 32 == 0 and b % 16 == 0
                    and _os.environ.get("DLAF_BT_KERNEL", "1") != "0"):
                R = -(-H // b) * b
                if R % 16 == 0:
                    from ..ops._ext import get_ext
                    Vp = torch.zeros((nwin, R, G), dtype=E.dtype, device=dev)
                    Vp[:, :H] = Vg
                    VTt = torch.zeros((nwin, G, R), dtype=E.dtype, device=dev)
                    VTt[:, :, :H] = torch.bmm(Vg, T).mT
                    if get_ext().bt_apply_group(Epad, Vp.contiguous(),
                                                VTt.contiguous(), 1 + s0,
                                                b, G, R, nwin):
                        continue
            # apply windows in ascending k (ordering constraint across
            # overlaps), per column strip on its own stream
            def _apply_strip(cs0, cs1):
                for k in range(nwin):
                    base = 1 + s0 + k * b
                    seg = Epad[base:base + H, cs0:cs1]
                    W = T[k] @ (Vg[k].mH @ seg)
                    seg -= Vg[k] @ W

            if not use_strips:
                _apply_strip(0, nE)
            else:
                ep = torch.cuda.Event()
                ep.record(streams[0])  # V/T prep ran on strip 0's stream
                for si in range(n_strips):
                    with torch.cuda.stream(streams[si]):
                        if si != 0:
                            streams[si].wait_event(ep)
                            # V/T were allocated on strip 0's stream; tell
                            # the caching allocator they are consumed here
                            Vg.record_stream(streams[si])
                            T.record_stream(streams[si])
                        _apply_strip(bounds[si], bounds[si + 1])
    if use_strips:
        cur = torch.cuda.current_stream(dev)
        for st_ in streams:
            cur.wait_stream(st_)
    E.copy_(Epad[:n])
