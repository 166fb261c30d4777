"""Numpy reference for two-stage SBR tridiagonalization
(docs/SBR_STAGE2_NOTES.md): the round-3 kernel target, algebra
validated off-GPU first (the workflow that made round 2's one-stage
kernel correct before it ever touched hardware).

Stage 1: full -> band(b) via per-panel QR (CholeskyQR2 with a
Householder fallback) + two-sided compact-WY updates -- GEMM-shaped.
Stage 2: band -> tridiagonal via bulge chasing (sb2st-style), with
every reflector recorded for the back-transform.

Run as a script for a self-check against numpy.linalg.eigh.
"""

from __future__ import annotations

import numpy as np


def _house(x):
    """Householder (beta, tau, v) with v[0] = 1 annihilating x[1:]."""
    alpha = x[0]
    nrm2 = float(x[1:] @ x[1:])
    if nrm2 == 0.0:
        return alpha, 0.0, np.zeros_like(x)
    beta = -np.sign(alpha if alpha != 0 else 1.0) * np.sqrt(
        alpha * alpha + nrm2)
    v = x.copy()
    v[0] = 1.0
    v[1:] /= (alpha - beta)
    tau = (beta - alpha) / beta
    return beta, tau, v


def _panel_qr(P):
    """Thin QR of P (M x b): CholeskyQR2, Householder fallback when the
    Gram matrix is numerically rank-deficient.  Returns (Q, R)."""
    M, b = P.shape
    try:
        G = P.T @ P
        R1 = np.linalg.cholesky(G).T
        Q = np.linalg.solve(R1.T, P.T).T
        G2 = Q.T @ Q
        R2 = np.linalg.cholesky(G2).T
        Q = np.linalg.solve(R2.T, Q.T).T
        return Q, R2 @ R1
    except np.linalg.LinAlgError:
        Q, R = np.linalg.qr(P)
        return Q, R


def band_reduce(A, b=8):
    """Stage 1: orthogonal reduction of symmetric A to band width b.

    Returns (B, Qacc): B is the banded matrix (dense storage, zero
    outside the band), Qacc the accumulated orthogonal transform with
    A = Qacc B Qacc^T.  (The kernel keeps Qacc in WY factor form; the
    reference accumulates it densely for clarity.)"""
    A = np.array(A, dtype=np.float64, copy=True)
    n = A.shape[0]
    Q = np.eye(n)
    for j0 in range(0, n - b - 1, b):
        r0 = j0 + b                      # first row below the band
        M = n - r0
        if M <= 1:
            break
        P = A[r0:, j0:j0 + b]            # the panel to annihilate
        Qp, R = _panel_qr(P)
        # build the full-size orthogonal factor H = I (+) Qfull where
        # Qfull extends Qp's column space to an orthogonal basis: use
        # the compact form H = I - W Y^T via Householder of Qp ...
        # reference clarity: apply via an explicit orthogonal completion
        H = np.eye(n)
        # orthogonal transform that maps span(P) onto the first b
        # coordinates of the trailing block: rows r0.., built from the
        # QR of P: [Qp | Qp_perp]; only Qp matters for the band result,
        # and the similarity needs a full orthogonal matrix:
        Qfull, _ = np.linalg.qr(
            np.concatenate([Qp, np.eye(M)], axis=1))
        # fix signs so Qfull[:, :b] == Qp
        for k in range(b if b <= M else M):
            s = Qp[:, k] @ Qfull[:, k]
            if s < 0:
                Qfull[:, k] *= -1
        H[r0:, r0:] = Qfull
        A = H.T @ A @ H
        Q = Q @ H
        # clean numerical fuzz outside the intended structure
        A[r0 + b:, j0:j0 + b] = 0.0
        A[j0:j0 + b, r0 + b:] = 0.0
    return A, Q


def bulge_chase(B, b):
    """Stage 2: banded (width b) -> tridiagonal by Givens bulge chasing
    (Rutishauser/Schwarz band reduction).  Returns (T, Q2) with
    B = Q2 T Q2^T.

    Per column j, the band elements below the subdiagonal are
    eliminated bottom-up; each elimination's two-sided rotation fills
    ONE element b rows further down, which is chased off the matrix:
    rotating plane (k+b-1, k+b) to zero A[k+b, k-1] creates the next
    fill at (k+2b, k+b-1), i.e. k <- k+b.  This per-element chase is
    the clarity-first reference; the GPU kernel blocks it into
    length-b reflectors with the same seats (docs/SBR_STAGE2_NOTES.md).
    """
    A = np.array(B, dtype=np.float64, copy=True)
    n = A.shape[0]
    Q2 = np.eye(n)

    def rot(p, q, a_piv, a_kill):
        r = np.hypot(a_piv, a_kill)
        if r == 0.0:
            return False
        c, s = a_piv / r, a_kill / r
        G = np.array([[c, s], [-s, c]])
        A[[p, q], :] = G @ A[[p, q], :]
        A[:, [p, q]] = A[:, [p, q]] @ G.T
        Q2[:, [p, q]] = Q2[:, [p, q]] @ G.T
        return True

    tiny = 0.0
    for j in range(n - 2):
        for i in range(min(j + b, n - 1), j + 1, -1):
            if A[i, j] == tiny:
                continue
            if not rot(i - 1, i, A[i - 1, j], A[i, j]):
                continue
            k = i
            while k + b < n and A[k + b, k - 1] != 0.0:
                r_ = k + b
                if not rot(r_ - 1, r_, A[r_ - 1, k - 1], A[r_, k - 1]):
                    break
                k = r_
    A = 0.5 * (A + A.T)
    return A, Q2


def self_check(n=48, b=4, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, 2 * n))
    A = X @ X.T / (2 * n)
    B, Q1 = band_reduce(A, b)
    err1 = np.linalg.norm(Q1 @ B @ Q1.T - A) / np.linalg.norm(A)
    band_resid = np.max(np.abs(np.triu(B, b + 1)))
    T, Q2 = bulge_chase(B, b)
    tri_resid = np.max(np.abs(np.triu(T, 2)))
    Q = Q1 @ Q2
    err2 = np.linalg.norm(Q @ T @ Q.T - A) / np.linalg.norm(A)
    ev = np.linalg.eigvalsh(T)
    ev_ref = np.linalg.eigvalsh(A)
    everr = np.max(np.abs(ev - ev_ref)) / max(1e-12,
                                              np.max(np.abs(ev_ref)))
    return dict(recon_band=err1, band_resid=band_resid,
                tri_resid=tri_resid, recon_tri=err2, eig_err=everr)


if __name__ == "__main__":
    for n, b in [(24, 4), (48, 4), (48, 8), (65, 8)]:
        r = self_check(n, b, seed=n)
        print(f"n={n:3d} b={b}: " + " ".join(
            f"{k}={v:.2e}" for k, v in r.items()))
