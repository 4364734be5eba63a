"""Divide & conquer tridiagonal eigensolver (Cuppen).

Counterpart of ``eigensolver/tridiag_solver/impl.h`` + ``merge.h`` (the
reference's flagship algorithm): recursive binary split, small dense leaves,
and rank-1 merges

    T = Qb (D + rho z z^T) Qb^H,  z = [Q1 last row, Q2 first row]/sqrt(2)

with (reference ``merge.h:1078-1214`` step structure):
  1. deflation (tiny z_i; близкие d pairs via Givens rotations on Q columns)
  2. secular-equation roots  1 + rho sum z_i^2/(d_i - λ) = 0 — here a
     VECTORIZED bracketed-Newton iteration over all roots simultaneously
     (device-resident), replacing the reference's per-root ``laed4`` host loop
  3. Gu/Eisenstat z-hat recomputation (orthogonality independent of root error)
  4. rank-1 eigenvectors + the big Qb @ U GEMM (device)

All O(k^2)+ work (secular iterations, U assembly, GEMMs) runs on the compute
device; only the O(k) deflation bookkeeping is host-side — the MI355X-native
replacement for the reference's bulk host thread teams.
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

_EPS = np.finfo(np.float64).eps


def _leaf(d: np.ndarray, e: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
    n = len(d)
    T = np.diag(d)
    if n > 1:
        T += np.diag(e, -1) + np.diag(e, 1)
    w, v = np.linalg.eigh(T)
    return w, v


def _secular_roots(d: torch.Tensor, z: torch.Tensor, rho: float,
                   iters: int = 40) -> Tuple[torch.Tensor, torch.Tensor]:
    """Roots of 1 + rho sum z_i^2/(d_i - lam), rho > 0, d ascending, z != 0.

    Returns (shift_idx [k] int64, mu [k]): lam_j = d[shift_idx_j] + mu_j; the
    (pole, offset) representation keeps d_i - lam_j accurate near poles.
    """
    k = d.shape[0]
    dev = d.device
    z2 = z * z
    if k == 1:
        return torch.zeros(1, dtype=torch.int64, device=dev), rho * z2
    if dev.type == "cuda" and d.dtype == torch.float64:
        # single-kernel path: one thread per root (csrc/secular.hip)
        from ..ops._ext import get_ext
        sidx = torch.empty(k, dtype=torch.int64, device=dev)
        mu = torch.empty(k, dtype=torch.float64, device=dev)
        get_ext().secular_roots(d.contiguous(), z2.contiguous(), float(rho), sidx, mu)
        return sidx, mu
    # interval per root j: (d_j, d_{j+1}), last: (d_{k-1}, d_{k-1} + rho)
    d_lo = d
    d_hi = torch.cat([d[1:], (d[-1] + rho).reshape(1)])
    mid = 0.5 * (d_lo + d_hi)
    # f(mid): [k] evaluations against all poles
    fm = 1.0 + rho * (z2.unsqueeze(1) / (d.unsqueeze(1) - mid.unsqueeze(0))).sum(0)
    # choose shift: left pole if f(mid) >= 0 (root left of mid), else right pole
    left = fm >= 0
    left = torch.cat([left[:-1], torch.ones(1, dtype=torch.bool, device=dev)])  # last root: left pole
    sidx = torch.where(left, torch.arange(k, device=dev),
                       torch.arange(1, k + 1, device=dev).clamp(max=k - 1))
    shift = d[sidx]
    # bracket for mu = lam - shift
    lo = torch.where(left, torch.zeros_like(d), mid - shift)
    hi = torch.where(left, mid - shift, torch.zeros_like(d))
    hi = torch.where(torch.arange(k, device=dev) == k - 1,
                     torch.full_like(d, rho), hi)
    delta0 = d.unsqueeze(1) - shift.unsqueeze(0)     # [k poles, k roots], exact
    mu = 0.5 * (lo + hi)
    # pole-split masks for the laed4-style two-pole rational model:
    # psi = poles i <= j, phi = poles i > j
    idx = torch.arange(k, device=dev)
    below = idx.unsqueeze(1) <= idx.unsqueeze(0)     # [pole i, root j]
    jj = idx                                          # left pole index = j
    j2 = (idx + 1).clamp(max=k - 1)                   # right pole (last: unused)
    last = idx == k - 1
    dscale = float(d.abs().max()) + rho
    for it in range(iters):
        diff = delta0 - mu.unsqueeze(0)              # d_i - lam_j
        t = z2.unsqueeze(1) / diff
        t2 = t / diff
        f = 1.0 + rho * t.sum(0)
        if it >= 12 and it % 6 == 0:
            # vectorized convergence check (one sync every 6 iterations)
            fp_est = rho * t2.sum(0)
            if float((f.abs() / fp_est.clamp_min(1e-300)).max()) < 1e-15 * dscale:
                break
        psi_p = torch.where(below, t2, torch.zeros_like(t2)).sum(0)
        phi_p = torch.where(below, torch.zeros_like(t2), t2).sum(0)
        d1 = delta0[jj, idx] - mu                    # d_j - lam (negative side)
        d2 = delta0[j2, idx] - mu                    # d_{j+1} - lam
        P = rho * psi_p * d1 * d1
        Q = rho * phi_p * d2 * d2
        c = f - rho * psi_p * d1 - rho * phi_p * d2
        # solve c + P/(d1 - s) + Q/(d2 - s) = 0 (interior);
        # quadratic a s^2 + b s + c2 = 0
        a = c
        b = -(c * (d1 + d2) + P + Q)
        c2 = c * d1 * d2 + P * d2 + Q * d1
        disc = (b * b - 4.0 * a * c2).clamp_min(0.0).sqrt()
        # stable quadratic roots; select the one inside the pole gap (d1, d2)
        qq = -0.5 * (b + torch.where(b >= 0, disc, -disc))
        safe_a = torch.where(a.abs() < 1e-300, torch.full_like(a, 1e-300), a)
        safe_q = torch.where(qq.abs() < 1e-300, torch.full_like(qq, 1e-300), qq)
        r1 = qq / safe_a
        r2 = c2 / safe_q
        in1 = (r1 > d1) & (r1 < d2)
        s_int = torch.where(in1, r1, r2)
        # last root: one-pole model c + P/(d1 - s) = 0
        s_last = d1 + P / torch.where(c.abs() < 1e-300, torch.full_like(c, 1e-300), c)
        s = torch.where(last, s_last, s_int)
        mu_n = mu + s
        lo = torch.where(f < 0, mu, lo)
        hi = torch.where(f > 0, mu, hi)
        bad = ~torch.isfinite(mu_n) | (mu_n <= lo) | (mu_n >= hi)
        mu = torch.where(bad, 0.5 * (lo + hi), mu_n)
    # Illinois (modified regula falsi) polish: the rational step can creep on
    # near-pole roots; Illinois has guaranteed bracket shrinkage and
    # superlinear convergence (f is monotone increasing in mu). Seed the
    # bracket with the rational iteration's (usually excellent) final point.
    big = torch.full_like(mu, 1e300)
    flo, fhi = -big, big
    diff = delta0 - mu.unsqueeze(0)
    fmu = 1.0 + rho * (z2.unsqueeze(1) / diff).sum(0)
    neg0 = fmu < 0
    lo = torch.where(neg0, mu, lo)
    flo = torch.where(neg0, fmu, flo)
    hi = torch.where(neg0, hi, mu)
    fhi = torch.where(neg0, fhi, fmu)
    side = torch.zeros_like(mu)           # -1 last update was lo, +1 was hi
    for _ in range(24):
        denom = fhi - flo
        x = torch.where(denom.abs() > 0, (lo * fhi - hi * flo) / denom,
                        0.5 * (lo + hi))
        inside = (x > lo) & (x < hi) & torch.isfinite(x)
        x = torch.where(inside, x, 0.5 * (lo + hi))
        diff = delta0 - x.unsqueeze(0)
        t = z2.unsqueeze(1) / diff
        fx = 1.0 + rho * t.sum(0)
        neg = fx < 0
        # Illinois halving when the same endpoint is kept twice
        fhi = torch.where(neg & (side < 0), 0.5 * fhi, fhi)
        flo = torch.where(~neg & (side > 0), 0.5 * flo, flo)
        lo = torch.where(neg, x, lo)
        flo = torch.where(neg, fx, flo)
        hi = torch.where(neg, hi, x)
        fhi = torch.where(neg, fhi, fx)
        side = torch.where(neg, -torch.ones_like(side), torch.ones_like(side))
        mu = x
    # final: return the bracket point with smaller |f|
    mu = torch.where(flo.abs() < fhi.abs(), lo, hi)
    return sidx, mu


def _merge(w1, Q1, w2, Q2, rho, device):
    """One rank-1 merge; inputs torch (device), returns (w, Q) device."""
    n1 = w1.shape[0]
    d = torch.cat([w1, w2])
    z = torch.cat([Q1[-1, :].conj(), Q2[0, :].conj()])
    n = d.shape[0]
    # normalize: T = D + rho z z^T, ||z||^2 = 2 -> z/=||z||, rho*=||z||^2
    znorm2 = float(z @ z)
    if znorm2 == 0 or rho == 0:
        Q = torch.zeros((n, n), dtype=Q1.dtype, device=device)
        Q[:n1, :n1] = Q1
        Q[n1:, n1:] = Q2
        w, idx = torch.sort(d)
        return w, Q[:, idx]
    rho_eff = rho * znorm2
    z = z / math.sqrt(znorm2)
    # reduce to rho > 0 by negation symmetry
    negate = rho_eff < 0
    if negate:
        d = -d
        rho_eff = -rho_eff
    # sort d ascending
    d_s, perm = torch.sort(d)
    z_s = z[perm]

    # ---- deflation (host bookkeeping, O(k), sequential scan in C++) ----
    dn = d_s.cpu().numpy().astype(np.float64).copy()
    zn = z_s.cpu().numpy().astype(np.float64).copy()
    k = n
    dmax = max(np.abs(dn).max(), rho_eff) if k else 1.0
    tol = 8.0 * _EPS * max(dmax, 1e-300)
    from ..ops._ext import get_ext
    dt_ = torch.from_numpy(dn)
    zt_ = torch.from_numpy(zn)
    deflated_t = torch.zeros(k, dtype=torch.bool)
    rots_t = torch.zeros((max(k, 1), 4), dtype=torch.float64)
    nrot = get_ext().dc_deflate_scan(dt_, zt_, float(rho_eff), float(tol),
                                     deflated_t, rots_t)
    deflated = deflated_t.numpy()
    rots = [(int(r[0]), int(r[1]), float(r[2]), float(r[3]))
            for r in rots_t[:nrot]]

    nd_idx = np.nonzero(~deflated)[0]
    df_idx = np.nonzero(deflated)[0]
    k1 = len(nd_idx)

    # ---- column assembly without materializing the full permuted Qb ----
    # (the reference's multiplyEigenvectors also avoids a dense re-layout,
    # merge.h:974-1076). Columns are gathered straight from the Q1/Q2 blocks;
    # rotation-involved columns are staged, rotated in order, and patched in.
    perm_np = perm.cpu().numpy()

    def gather_cols(cols_np):
        m = len(cols_np)
        out = torch.zeros((n, m), dtype=Q1.dtype, device=device)
        if m == 0:
            return out
        pc = perm_np[cols_np]
        top = pc < n1
        ti = np.nonzero(top)[0]
        bi = np.nonzero(~top)[0]
        if len(ti):
            out[:n1, torch.from_numpy(ti).to(device)] = \
                Q1[:, torch.from_numpy(pc[ti]).to(device)]
        if len(bi):
            out[n1:, torch.from_numpy(bi).to(device)] = \
                Q2[:, torch.from_numpy(pc[bi] - n1).to(device)]
        return out

    rot_cols = sorted({c for (i, j, _, _) in rots for c in (i, j)})
    rot_pos = {c: p for p, c in enumerate(rot_cols)}
    R = gather_cols(np.array(rot_cols, dtype=np.int64))
    # R <- R G per rotation, G = [[c, -s], [s, c]] (z' = G^T z zeroes
    # component i). Rotations sharing a column must apply in order, but
    # disjoint ones commute: greedy-schedule into ROUNDS of column-disjoint
    # pairs and apply each round as one batched 4-kernel update (the
    # per-rotation loop was 4 tiny device launches each; same class of
    # host-bound overhead as the bt prep fix, profiles/
    # bench_full_r2_kernel_stats.md).
    import os as _os
    if _os.environ.get("DLAF_DC_ROT_BATCH", "0") != "0":
        rounds: list = []
        last: dict = {}
        for rot in rots:
            i, j = rot[0], rot[1]
            r = max(last.get(i, 0), last.get(j, 0))
            if r == len(rounds):
                rounds.append([])
            rounds[r].append(rot)
            last[i] = last[j] = r + 1
        rdt = R.real.dtype if R.is_complex() else R.dtype
        for rnd in rounds:
            pi = torch.tensor([rot_pos[i] for (i, _, _, _) in rnd],
                              dtype=torch.int64, device=device)
            pj = torch.tensor([rot_pos[j] for (_, j, _, _) in rnd],
                              dtype=torch.int64, device=device)
            cc = torch.tensor([c for (_, _, c, _) in rnd], dtype=rdt,
                              device=device)
            ss = torch.tensor([s for (_, _, _, s) in rnd], dtype=rdt,
                              device=device)
            gi = R[:, pi]
            gj = R[:, pj]
            R[:, pi] = cc * gi + ss * gj
            R[:, pj] = -ss * gi + cc * gj
    else:
        for (i, j, c, s) in rots:
            pi_, pj_ = rot_pos[i], rot_pos[j]
            gi = R[:, pi_].clone()
            gj = R[:, pj_].clone()
            R[:, pi_] = c * gi + s * gj
            R[:, pj_] = -s * gi + c * gj

    def patch_rotated(out, cols_np):
        pos = [(p, rot_pos[c]) for p, c in enumerate(cols_np) if c in rot_pos]
        if pos:
            dst = torch.tensor([p for p, _ in pos], dtype=torch.int64, device=device)
            src = torch.tensor([q for _, q in pos], dtype=torch.int64, device=device)
            out[:, dst] = R[:, src]
        return out

    Qnd = patch_rotated(gather_cols(nd_idx), nd_idx)
    Qdf = patch_rotated(gather_cols(df_idx), df_idx)

    if k1 > 0:
        # the HIP one-thread-per-root kernel handles every size in one
        # launch; the torch formulation stays for CPU runs (small merges on
        # host tensors, large on device)
        if device.type == "cuda" and d_s.dtype == torch.float64:
            sec_dev = device
        else:
            sec_dev = device if k1 >= 192 else torch.device("cpu")
        dk = torch.from_numpy(dn[nd_idx]).to(sec_dev)
        zk = torch.from_numpy(zn[nd_idx]).to(sec_dev)
        sidx, mu = _secular_roots(dk, zk, rho_eff)
        lam = dk[sidx] + mu
        # delta[i, j] = d_i - lam_j, via exact pole differences
        delta = (dk.unsqueeze(1) - dk[sidx].unsqueeze(0)) - mu.unsqueeze(0)
        # Gu/Eisenstat z-hat:
        #   zh_i^2 = prod_j (lam_j - d_i) / (rho * prod_{j!=i} (d_j - d_i))
        dd = dk.unsqueeze(1) - dk.unsqueeze(0)       # d_j - d_i at [i, j]
        num = (-delta).abs().clamp_min(1e-300).log().sum(1)
        den_m = dd.abs().clamp_min(1e-300).log()
        den = den_m.sum(1) - torch.diagonal(den_m)
        zh = torch.exp(0.5 * (num - den - math.log(rho_eff)))
        zh = torch.where(zk < 0, -zh, zh)
        # eigenvectors of the rank-1 system
        U = zh.unsqueeze(1) / delta                  # [k1, k1]
        U = U / torch.linalg.vector_norm(U, dim=0, keepdim=True)
        lam_out = lam.to(device)
        V_nd = Qnd @ U.to(Qnd.dtype).to(device)
    else:
        lam_out = torch.empty(0, dtype=d_s.dtype, device=device)
        V_nd = torch.empty((n, 0), dtype=Qnd.dtype, device=device)

    all_vals = torch.cat([lam_out, torch.from_numpy(dn[df_idx]).to(device)])
    if negate:
        all_vals = -all_vals
    order = torch.argsort(all_vals)
    w_out = all_vals[order]
    # scatter the nd / deflated columns straight to their final positions
    invp = torch.empty(n, dtype=torch.int64, device=device)
    invp[order] = torch.arange(n, device=device)
    Q_out = torch.empty((n, n), dtype=Qnd.dtype, device=device)
    if k1:
        Q_out.index_copy_(1, invp[:k1], V_nd)
    if n - k1:
        Q_out.index_copy_(1, invp[k1:], Qdf)
    return w_out, Q_out


def tridiagonal_eigensolver(d: torch.Tensor, e: torch.Tensor,
                            device=None, leaf: int = None):
    """Eigendecomposition of a real symmetric tridiagonal matrix.

    Returns (evals [n] fp, evecs [n, n]) on ``device``. Reference:
    ``eigensolver/tridiag_solver/impl.h:198-278`` (local).
    """
    if device is not None and not isinstance(device, torch.device):
        device = torch.device(device)
    if leaf is None:
        import os
        leaf = int(os.environ.get("DLAF_DC_LEAF", "64"))
    if device is None:
        device = d.device
    dn = d.detach().cpu().numpy().astype(np.float64).copy()
    en = e.detach().cpu().numpy().astype(np.float64).copy()
    n = len(dn)
    if n == 0:
        z = torch.zeros(0, dtype=torch.float64, device=device)
        return z, torch.zeros((0, 0), dtype=torch.float64, device=device)

    def solve(lo: int, hi: int):
        m = hi - lo
        if m <= leaf:
            w, v = _leaf(dn[lo:hi].copy(), en[lo:hi - 1].copy())
            return (torch.from_numpy(w).to(device), torch.from_numpy(v).to(device))
        mid = lo + m // 2
        rho = float(en[mid - 1])
        # Cuppen: subtract the coupling from the two touching diagonals
        dn[mid - 1] -= rho
        dn[mid] -= rho
        w1, Q1 = solve(lo, mid)
        w2, Q2 = solve(mid, hi)
        return _merge(w1, Q1, w2, Q2, rho, device)

    return solve(0, n)
