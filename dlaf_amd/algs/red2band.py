"""Reduction to band form (first stage of the two-stage eigensolver) and its
back-transform.

Counterpart of ``eigensolver/reduction_to_band/impl.h`` (local 993-1149) and
``eigensolver/bt_reduction_to_band/impl.h``: A (Hermitian, lower) is reduced
panel-by-panel to a Hermitian band matrix of bandwidth ``band`` by blocked
Householder transforms,

    A_band = Q^H A Q,   Q = Q_p1 Q_p2 ... ,   Q_pk = I - V_k T_k V_k^H

with the reflectors V stored LAPACK-style below the R block of each panel and
the band (diagonal blocks + upper-triangular subdiagonal R blocks) left in
place.

MI355X-native design notes: the panel QR is a thin column loop (larfg +
rank-1 update, device-resident); everything O(n^2 b) and above — T factors
(V^H V + one triangular inverse), the trailing two-sided update
(A22 -= V X^H + X V^H with X = A22 V T - 1/2 V (T^H V^H A22 V T)) and the
back-transform E <- Q E — is plain large GEMMs. This v1 operates on a dense
device-resident Hermitian matrix (the tiled Matrix is packed/unpacked at the
boundary); the distributed panel/trailing variant follows the Cholesky-style
panel broadcast machinery.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..matrix.matrix import Matrix
from ..ops import tile_ops as ops


def _herm_full_dense(a: torch.Tensor) -> torch.Tensor:
    return torch.tril(a) + torch.tril(a, -1).mH


def larfg(x: torch.Tensor) -> Tuple[torch.Tensor, complex]:
    """LAPACK-convention Householder generator, in place (host-sync variant).

    x[0] <- beta (real), x[1:] <- v tail (v0 = 1 implicit); returns tau.
    H = I - tau v v^H with H^H x = beta e1.
    """
    alpha = x[0].clone()
    tail = x[1:]
    xnorm = torch.linalg.vector_norm(tail).item() if tail.numel() else 0.0
    a_re = alpha.real.item() if x.is_complex() else alpha.item()
    a_im = alpha.imag.item() if x.is_complex() else 0.0
    if xnorm == 0.0 and a_im == 0.0:
        return x[0], 0.0  # H = I
    beta = -((a_re * a_re + a_im * a_im + xnorm * xnorm) ** 0.5)
    if a_re < 0:
        beta = -beta
    # LAPACK zlarfg: tau = (beta - alpha)/beta, v = x / (alpha - beta), beta real
    tau = complex((beta - a_re) / beta, -a_im / beta) if x.is_complex() else (beta - a_re) / beta
    tail.div_(alpha - beta)
    x[0] = beta
    return x[0], tau


def _larfg_device(x: torch.Tensor, taus: torch.Tensor, j: int) -> torch.Tensor:
    """Branchless device larfg: no host synchronization.

    Writes beta into x[0], v-tail into x[1:], tau into taus[j]; returns tau as
    a 0-dim device tensor. Degenerate columns (nothing to eliminate) get
    tau = 0 and x untouched, matching LAPACK.
    """
    alpha = x[0].clone()
    tail = x[1:]
    n2 = (tail.conj() * tail).sum().real if x.is_complex() else (tail * tail).sum()
    a_re = alpha.real if x.is_complex() else alpha
    a_im = alpha.imag if x.is_complex() else torch.zeros_like(a_re)
    degen = (n2 == 0) & (a_im == 0)
    mag = torch.sqrt(a_re * a_re + a_im * a_im + n2)
    beta = torch.where(a_re < 0, mag, -mag)
    safe_beta = torch.where(degen, torch.ones_like(beta), beta)
    if x.is_complex():
        tau = torch.complex((safe_beta - a_re) / safe_beta, -a_im / safe_beta)
        denom = alpha - safe_beta.to(alpha.dtype)
    else:
        tau = (safe_beta - a_re) / safe_beta
        denom = alpha - safe_beta
    tau = torch.where(degen, torch.zeros_like(tau), tau)
    scale = torch.where(degen, torch.ones_like(denom), 1.0 / denom)
    tail.mul_(scale)
    x[0] = torch.where(degen, alpha, beta.to(alpha.dtype))
    taus[j] = tau
    return tau


def panel_qr_(P: torch.Tensor, taus: torch.Tensor) -> None:
    """In-place unblocked QR of a tall panel, LAPACK geqrf convention.

    R in the upper triangle, reflector tails below the diagonal (unit
    implicit), taus filled. GPU: ONE cooperative kernel launch for the whole
    panel (csrc/panel_qr.hip — the column loop with its grid syncs runs on
    device). CPU: torch column loop (the reference backend).
    """
    m, nb = P.shape
    if P.is_cuda and P.stride(1) == 1 and m > 0 and nb > 0:
        from ..ops._ext import get_ext
        rdt = torch.float32 if P.dtype in (torch.float32, torch.complex64) \
            else torch.float64
        norms = torch.zeros(nb, dtype=rdt, device=P.device)
        wraw = torch.zeros(3 * nb * nb, dtype=P.dtype, device=P.device)
        get_ext().panel_qr(P, taus, norms, wraw)
        return
    one = torch.ones(1, dtype=P.dtype, device=P.device)
    for j in range(min(m, nb)):
        tau = _larfg_device(P[:, j] if j == 0 else P[j:, j], taus, j)
        if j + 1 < nb:
            # apply H^H = I - conj(tau) v v^H from the left (zgeqr2 convention)
            v = torch.cat([one, P[j + 1:, j]])
            w = v.conj() @ P[j:, j + 1:]
            P[j:, j + 1:] -= (tau.conj() * torch.outer(v, w))


def t_factor(V: torch.Tensor, taus: torch.Tensor) -> torch.Tensor:
    """Compact-WY T (upper triangular): Q = I - V T V^H.

    T = inv(diag(1/tau) + striu(V^H V)); rows/cols with tau = 0 vanish.
    One Gram GEMM + one small triangular inverse (reference larft analog,
    ``factorization/qr/t_factor_impl.h``).
    """
    G = V.mH @ V
    zc = taus == 0
    safe = torch.where(zc, torch.ones_like(taus), taus)
    M = torch.triu(G, 1) + torch.diag(1.0 / safe)
    M = torch.where(zc[None, :] | zc[:, None], torch.zeros_like(M), M)
    M = M + torch.diag(torch.where(zc, torch.ones_like(safe), torch.zeros_like(safe)))
    T = ops.tri_inverse_full(M.contiguous(), lower=False)
    T = torch.where(zc[None, :] | zc[:, None], torch.zeros_like(T), T)
    return T


def reduction_to_band_dense(A: torch.Tensor, band: int):
    """Reduce dense Hermitian A (full storage) to band form in place.

    Returns (taus, panels) where panels is a list of (j0, nrefl) descriptors;
    reflectors live in A[j0+band:, j0:j0+band] strictly below the R block.
    """
    n = A.shape[0]
    dt, dev = A.dtype, A.device
    taus_all = []
    panels = []
    for j0 in range(0, max(n - band, 0), band):
        r0 = j0 + band
        bw = min(band, n - j0 - band)  # panel width
        if bw <= 0:
            break
        P = A[r0:, j0:j0 + bw]
        m_p = P.shape[0]
        nrefl = min(m_p, bw)
        taus = torch.zeros(nrefl, dtype=dt, device=dev)
        panel_qr_(P, taus)
        panels.append((j0, bw, nrefl))
        taus_all.append(taus)
        # V: unit lower-trapezoidal from P
        V = torch.tril(P[:, :nrefl], -1)
        V = V + torch.eye(m_p, nrefl, dtype=dt, device=dev)
        T = t_factor(V, taus)
        if bw < band:
            # capped panel (n - j0 - band < band): the remaining band columns
            # [j0+bw, j0+band) still have rows >= r0 inside the band and must
            # receive the LEFT factor Q^H = I - V T^H V^H (they are not part
            # of the trailing two-sided update below)
            Bblk = A[r0:, j0 + bw:j0 + band]
            Wb = T.mH @ (V.mH @ Bblk)
            Bblk.addmm_(V, Wb, beta=1, alpha=-1)
            A[j0 + bw:j0 + band, r0:] = Bblk.mH
        A22 = A[r0:, r0:]
        Y = A22 @ (V @ T)                      # hemm (A22 kept full Hermitian)
        S = T.mH @ (V.mH @ Y)
        X = Y - 0.5 * V @ S
        # in-place rank-2b update: no m_t x m_t temporaries
        A22.addmm_(V, X.mH, beta=1, alpha=-1)
        A22.addmm_(X, V.mH, beta=1, alpha=-1)
    return taus_all, panels


def reduction_to_band(mat: Matrix, band: Optional[int] = None, grid=None):
    """Reduce a Hermitian tiled Matrix (lower) to band form; returns taus.

    ``mat`` is overwritten: band in place, reflectors below (LAPACK-style).
    Reference: ``eigensolver/reduction_to_band/impl.h:993-1149``.
    """
    d = mat.dist
    assert d.m == d.n and d.mb == d.nb
    if band is None:
        band = d.nb
    # any band >= 1 works on the dense local formulation (the tiled
    # DISTRIBUTED panels require nb % band == 0; see eigensolver_tiled)
    g = grid if grid is not None else mat.grid
    assert g is None or not g.distributed, \
        "distributed reduction_to_band lands with the distributed eigensolver"
    A = _herm_full_dense(mat.to_global())
    taus_all, panels = reduction_to_band_dense(A, band)
    mat.set_from_global(A)
    return {"taus": taus_all, "panels": panels, "band": band}


def bt_reduction_to_band(E: torch.Tensor, mat_v: Matrix, refl) -> None:
    """Back-transform E <- Q E (dense device E), panels applied in reverse.

    Reference: ``eigensolver/bt_reduction_to_band/impl.h:1-399``.
    """
    band = refl["band"]
    A = mat_v.to_global()
    n = A.shape[0]
    for (j0, bw, nrefl), taus in zip(reversed(refl["panels"]), reversed(refl["taus"])):
        r0 = j0 + band
        P = A[r0:, j0:j0 + nrefl]
        m_p = P.shape[0]
        V = torch.tril(P, -1) + torch.eye(m_p, nrefl, dtype=A.dtype, device=A.device)
        T = t_factor(V, taus)
        W = T @ (V.mH @ E[r0:, :])
        E[r0:, :].addmm_(V, W, beta=1, alpha=-1)
