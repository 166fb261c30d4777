"""Perturbative warm-started eigendecomposition tracking (EXPERIMENTAL,
opt-in via ``KFAC_EIG_TRACKER=1``).

K-FAC recomputes each factor's eigendecomposition every
``kfac_update_freq`` steps.  When the factor moves slowly between
updates (large effective batch, large im2col row counts, or
``factor_decay`` close to 0), the new eigenbasis is a small rotation of
the previous one and a full library eigensolve is waste.  This tracker:

1. rotates the new factor into the previous basis ``B = Q^T A Q``
   (two MFMA-backed fp32 GEMMs);
2. applies the first-order (Rayleigh-Schroedinger) eigenvector
   correction ``Q <- Q (I + S)``, ``S_ij = B_ij / (d_j - d_i)``,
   **gap-gated**: entries with gaps below ``gap_rel*(|d_i|+|d_j|)`` are
   zeroed -- near-degenerate clusters are deliberately left mixed (the
   K-FAC denominator ``dG_i dA_j + damping`` is insensitive to basis
   rotation inside a near-equal eigenvalue cluster);
3. one Newton-Schulz step restores fp32 orthogonality;
4. eigenvalues get the standard second-order correction
   ``d_i - sum_j B_ij S_ij``.

Safety: the caller runs the two-phase protocol (:func:`tracked_eig_multi`)
-- ``prepare`` computes B plus two scalar health stats per factor
(gated residual mass and ||S||), ONE batched host transfer reads all
stats, and factors whose stats exceed tolerance take a cold library
solve instead of committing the warm update.  At the reference's
default ``factor_decay=0.95`` the factor is 95% fresh batch noise per
step, so small-sample layers cold-restart nearly always -- which is why
this path is opt-in; the default production path is the async/batched
rocSOLVER tier in ``linalg.mat_eig_multi``.

Reference being replaced: the serial per-layer eigensolve loop
(kfac/kfac_preconditioner_eigen.py:98-119).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

__all__ = ["EigenTracker", "tracked_eig_multi"]


class EigenTracker:
    def __init__(self, cold_every: int = 50, cold_tol: float = 0.1,
                 s_tol: float = 0.15, gap_rel: float = 0.03,
                 gap_floor_rel: float = 1e-5, gap_abs: float = 1e-12):
        # cold_tol: sqrt(gated off-mass / total mass) of B the correction
        #   is asked to remove; beyond it first-order is invalid.
        # s_tol: RMS of (clamped) S; beyond it (I+S) strays too far from
        #   orthogonal for two Newton-Schulz steps.
        # gap_rel: mixing tolerance.  Rotating eigenvectors whose
        #   eigenvalues differ by Delta changes the K-FAC denominator
        #   1/(dG*dA + damping) by at most Delta/(d + damping/dG_max)
        #   relative -- a RELATIVE gap criterion; leaving pairs with
        #   <= 3% eigenvalue difference mixed bounds the preconditioner
        #   error at ~3%, far below the statistical noise of the factor
        #   estimate itself.
        # gap_floor_rel: absolute floor as a fraction of d_max -- the
        #   near-zero bulk of rank-deficient sample covariances is one
        #   cluster (also beneath fp32 eigensolver resolution).
        self.cold_every = cold_every
        self.cold_tol = cold_tol
        self.s_tol = s_tol
        self.gap_rel = gap_rel
        self.gap_floor_rel = gap_floor_rel
        self.gap_abs = gap_abs
        self.Q: Optional[torch.Tensor] = None
        self.calls_since_cold = 0
        self.cold_count = 0
        self.warm_count = 0
        self._pending = None

    # -- cold path ----------------------------------------------------------
    def needs_cold(self, A: torch.Tensor) -> bool:
        return (self.Q is None or self.Q.shape[-1] != A.shape[-1]
                or self.calls_since_cold >= self.cold_every)

    def seed(self, w: torch.Tensor, Q: torch.Tensor) -> None:
        self.Q = Q
        self.calls_since_cold = 0
        self.cold_count += 1

    def _cold(self, A: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        from kfac_pytorch_amd.ops.linalg import mat_eig
        w, Q = mat_eig(A, method="eigh")
        self.seed(w, Q.clone())
        return w, Q

    # -- warm path, two-phase ----------------------------------------------
    def prepare(self, A: torch.Tensor) -> torch.Tensor:
        """Queue the warm-update GPU work; returns a (2,) stats tensor
        [gated_residual_rel, S_rms] (GPU, no sync)."""
        Q = self.Q
        B = Q.t() @ (A @ Q)
        B = 0.5 * (B + B.t())
        d = B.diagonal()
        gap = d.unsqueeze(0) - d.unsqueeze(1)      # gap[i,j] = d_j - d_i
        thr = (self.gap_rel * (d.abs().unsqueeze(0) + d.abs().unsqueeze(1))
               + self.gap_floor_rel * d.abs().max()
               + self.gap_abs)
        mask = gap.abs() > thr
        S = torch.where(mask,
                        B / torch.where(mask, gap, torch.ones_like(gap)),
                        torch.zeros_like(B))
        S.fill_diagonal_(0.0)
        # damped-Jacobi style clamp: oversized first-order rotations are
        # applied partially (still reduces off-mass; keeps I+S in the
        # Newton-Schulz convergence basin)
        S.clamp_(-0.25, 0.25)
        m = A.shape[-1]
        off2 = (B * B * mask.to(B.dtype)).sum()
        tot2 = (B * B).sum().clamp_min(1e-30)
        stats = torch.stack([(off2 / tot2).sqrt(),
                             S.norm() / (m ** 0.5)])
        self._pending = (B, d, S)
        return stats

    def apply_(self) -> Tuple[torch.Tensor, torch.Tensor]:
        """Apply the prepared (clamped) rotation to the tracked basis;
        returns the corrected (eigenvalues, Q).  Does NOT touch the
        warm/cold counters -- callers iterate this until the health
        stats go green (quadratic convergence near the solution, the
        clamp makes distant starts contract too)."""
        B, d, S = self._pending
        self._pending = None
        Q = self.Q
        Qn = Q + Q @ S
        for _ in range(2):  # Newton-Schulz: error cubes per step
            G = Qn.t() @ Qn
            Qn = 1.5 * Qn - 0.5 * (Qn @ G)
        self.Q = Qn
        d_corr = d - (B * S).sum(dim=1)
        return d_corr, Qn

    def commit(self, A: torch.Tensor,
               ok: bool) -> Tuple[torch.Tensor, torch.Tensor]:
        """Finalize one tracked update: apply if ``ok``, else cold."""
        if not ok:
            self._pending = None
            return self._cold(A)
        self.calls_since_cold += 1
        self.warm_count += 1
        return self.apply_()

    def is_green(self, stats) -> bool:
        rel, s_rms = float(stats[0]), float(stats[1])
        return rel <= self.cold_tol and s_rms <= self.s_tol

    def update(self, A: torch.Tensor, max_rounds: int = 4
               ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Single-factor convenience wrapper (one host sync per round)."""
        if self.needs_cold(A):
            return self._cold(A)
        result = None
        for _ in range(max_rounds):
            stats = self.prepare(A).cpu()
            green = self.is_green(stats)
            result = self.apply_()
            if green:
                self.calls_since_cold += 1
                self.warm_count += 1
                return result
        return self._cold(A)


def tracked_eig_multi(trackers: List["EigenTracker"], mats,
                      max_rounds: int = 4) -> list:
    """Eigendecompose many factors with warm ITERATIVE tracking: each
    round queues every active factor's correction on-stream, ONE host
    transfer reads all health stats, green factors finish, red ones
    iterate (refined basis) up to ``max_rounds``; whatever is left joins
    the batched cold solve.  Returns [(w, Q), ...] aligned with
    ``mats``."""
    import os
    from kfac_pytorch_amd.ops.linalg import mat_eig_multi
    debug = os.environ.get("KFAC_TRACKER_DEBUG")
    out = [None] * len(mats)
    cold = [i for i, t in enumerate(trackers) if t.needs_cold(mats[i])]
    active = [i for i in range(len(mats)) if i not in set(cold)]

    for rnd in range(max_rounds):
        if not active:
            break
        stats = [trackers[i].prepare(mats[i]) for i in active]
        host = torch.stack(stats).cpu()
        if debug:
            print(f"tracker round {rnd} (m, rel, s_rms, green):",
                  [(int(mats[i].shape[-1]), round(float(host[k, 0]), 4),
                    round(float(host[k, 1]), 4),
                    trackers[i].is_green(host[k]))
                   for k, i in enumerate(active)], flush=True)
        still = []
        for k, i in enumerate(active):
            tr = trackers[i]
            green = tr.is_green(host[k])
            hopeless = (float(host[k, 0]) > 4 * tr.cold_tol
                        or float(host[k, 1]) > 4 * tr.s_tol)
            if hopeless and not green:
                # basis moved too far for iterative refinement to pay
                # off -- skip the wasted rounds, go straight to cold
                tr._pending = None
                cold.append(i)
                continue
            result = tr.apply_()
            if green:
                tr.calls_since_cold += 1
                tr.warm_count += 1
                out[i] = result
            else:
                still.append(i)
        active = still

    cold.extend(active)  # unconverged after max_rounds
    if cold:
        solved = mat_eig_multi([mats[i] for i in cold], need_sorted=False)
        for i, (w, Q) in zip(cold, solved):
            trackers[i]._pending = None
            trackers[i].seed(w, Q.clone())
            out[i] = (w, Q)
    return out
