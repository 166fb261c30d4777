"""Reference implementation of the batched blocked tridiagonalization
(the algorithm behind ops/csrc_solver/sytrd_panel.hip), in numpy.

This pins the exact storage conventions and the deferred-alpha panel
algebra the HIP kernel uses, and is the CPU oracle for
tests/test_sytrd_ref.py.  It replaces the contract of rocSOLVER's
``ssytrd`` (the reference framework's eigensolve goes through
cuSOLVER ``cusolverDnSsyevd``, /root/reference/packages/tcmm/src/
tcmm_kernel.cu:56-116; its tridiagonalization stage is ~80% of the
solve and the round-2 custom-kernel target).

Conventions (row-major A, symmetric input, all 0-indexed):

* column step j (j = 0..n-2) builds a Householder reflector that
  annihilates row j's elements beyond j+1:  v_j has an implicit-unit
  at r = j+1 and values s*x at r >= j+2;
* the buffer is the torch row-major tensor; read COLUMN-major by
  LAPACK/rocSOLVER it is exactly the uplo=LOWER sytrd output format
  (row-major row i == column-major column i for the symmetric input),
  so rocsolver_sstedc / sormtr and the WY back-transform can consume
  it directly;
* within a panel, W columns are stored PRE-alpha (w1 = tau*(A v -
  corrections)) and every use expands the alpha fix-up into per-column
  scalar coefficients -- this is what lets the GPU kernel run one
  grid-barrier phase for partials and one for the matvec, instead of
  three;
* after the panel, W is finalized (w1 + alpha*v) and the trailing
  full square gets the rank-2*ib update via two GEMMs (full symmetric
  storage is kept so the per-column matvec streams contiguous rows).

On exit: A holds v values in row i at columns >= i+2 with the UNIT
stored explicitly at column i+1 (LAPACK keeps E there instead; nothing
downstream reads that slot), the diagonal holds D, and (D, E, tau) are
returned separately.
"""

from __future__ import annotations

import numpy as np


def _larfg(x1: float, xs: np.ndarray):
    """Householder generation on [x1; xs]: returns (beta, tau, s) with
    v = [1; s*xs] such that (I - tau v v^T)[x1; xs] = [beta; 0]."""
    nrm2 = float(xs @ xs)
    if nrm2 == 0.0:
        return x1, 0.0, 0.0
    beta = -np.sign(x1 if x1 != 0 else 1.0) * np.sqrt(x1 * x1 + nrm2)
    tau = (beta - x1) / beta
    s = 1.0 / (x1 - beta)
    return beta, tau, s


def sytrd_blocked(A: np.ndarray, nb: int = 64):
    """Blocked tridiagonalization, deferred-alpha panel algebra.

    Returns (A, D, E, tau): A is overwritten as described above.
    """
    A = np.array(A, dtype=np.float64, copy=True)
    n = A.shape[0]
    D = np.zeros(n)
    E = np.zeros(n)
    tau = np.zeros(n)
    for j0 in range(0, n - 1, nb):
        ib = min(nb, n - 1 - j0)
        W1 = np.zeros((ib, n))          # pre-alpha W columns, as rows
        alpha = np.zeros(ib)
        pwv_prev = 0.0
        tau_prev = 0.0
        for i in range(ib):
            j = j0 + i
            if i > 0:
                alpha[i - 1] = -0.5 * tau_prev * pwv_prev
            # ---- phase A: correct x = A[j, j:] with deferred-alpha W
            x = A[j, j:].copy()         # elements r in [j, n)
            for c in range(i):
                vc = A[j0 + c, j:]      # scaled v values (unit at its
                                        # own sub-diagonal, already
                                        # stored by its column's pass)
                cb = A[j0 + c, j]       # V_c(j)
                coefA = W1[c, j] + 2.0 * alpha[c] * cb
                x -= vc * coefA + W1[c, j:] * cb
            A[j, j:] = x                # pre-scale writeback
            # partials the kernel reduces across workgroups:
            # nrm2 over r >= j+2, x1 = x[j+1]
            x1 = x[1]
            beta, tau_j, s = _larfg(x1, x[2:])
            E[j] = beta
            tau[j] = tau_j
            # scaled-v writeback (the kernel does this lazily in the
            # NEXT column's phase A / the finalize pass; numerically
            # identical)
            A[j, j + 1] = 1.0
            A[j, j + 2:] = s * x[2:]
            v = A[j, j + 1:]            # the reflector, unit explicit
            # ---- phase B: matvec with pre-scale x + scalar folding
            # A.v restricted to r in (j, n):
            w_pre = A[j + 1:, j + 1] * 1.0 + s * (A[j + 1:, j + 2:] @ x[2:])
            for c in range(i):
                vc = A[j0 + c, j + 1:]
                sV = float(vc @ v)      # kernel: from pV partial + fixup
                sW = float(W1[c, j + 1:] @ v)
                w_pre -= vc * (sW + 2.0 * alpha[c] * sV) \
                    + W1[c, j + 1:] * sV
            w1 = tau_j * w_pre
            W1[i, j + 1:] = w1
            pwv_prev = float(w1 @ v)
            tau_prev = tau_j
        # ---- finalize W (apply alphas) and the trailing update
        alpha[ib - 1] = -0.5 * tau_prev * pwv_prev
        for c in range(ib):
            jc = j0 + c
            W1[c, jc + 1:] += alpha[c] * A[jc, jc + 1:]
        t = j0 + ib
        if t < n:
            V2 = A[j0:j0 + ib, t:]      # (ib, n-t): v values, rows
            W2 = W1[:, t:]
            A[t:, t:] -= V2.T @ W2 + W2.T @ V2
    for j in range(n):
        D[j] = A[j, j]
    E[n - 1] = 0.0
    # E as LAPACK defines it: E[j] = subdiagonal j, j = 0..n-2; our loop
    # covered j = 0..n-2 (the last one via the degenerate larfg).
    return A, D, E, tau


def build_q(A: np.ndarray, tau: np.ndarray):
    """Accumulate Q = H_0 H_1 ... H_{n-2} from the stored reflectors
    (row i of A holds v_i at columns >= i+1 with the unit explicit)."""
    n = A.shape[0]
    Q = np.eye(n)
    for j in range(n - 2, -1, -1):
        v = np.zeros(n)
        v[j + 1:] = A[j, j + 1:]
        Q -= tau[j] * np.outer(v, v @ Q)
    return Q


def tridiag(D: np.ndarray, E: np.ndarray):
    n = D.shape[0]
    T = np.diag(D)
    for j in range(n - 1):
        T[j, j + 1] = T[j + 1, j] = E[j]
    return T


if __name__ == "__main__":
    rng = np.random.default_rng(0)
    for n, nb in [(8, 4), (37, 8), (129, 64), (200, 64), (256, 32)]:
        X = rng.standard_normal((n, 2 * n))
        A = X @ X.T / (2 * n)
        Aout, D, E, tau = sytrd_blocked(A, nb)
        Q = build_q(Aout, tau)
        T = tridiag(D, E)
        rec = np.linalg.norm(Q @ T @ Q.T - A) / np.linalg.norm(A)
        ev = np.linalg.eigvalsh(T)
        ev_ref = np.linalg.eigvalsh(A)
        everr = np.max(np.abs(ev - ev_ref)) / max(1e-12,
                                                  np.max(np.abs(ev_ref)))
        orth = np.linalg.norm(Q @ Q.T - np.eye(n))
        print(f"n={n:4d} nb={nb:3d} rec={rec:.2e} ev={everr:.2e} "
              f"orth={orth:.2e}")
        assert rec < 1e-12 and everr < 1e-11 and orth < 1e-12, (n, nb)
    print("sytrd_ref: all checks passed")
