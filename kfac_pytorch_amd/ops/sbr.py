"""Two-stage SBR tridiagonalization, stage 1: batched full -> band
reduction at GEMM rate (docs/SBR_STAGE2_NOTES.md; numpy oracle
scripts/sbr_ref.py, validated to machine precision).

This is the round-3 eigensolver path's first half, shipped as a
working, tested torch implementation: every panel is annihilated with
ONE batched ``geqrf`` (no per-column sequencing -- the measured cost
of every one-stage tridiagonalization, docs/SYTRD_DESIGN.md), and the
two-sided update runs as four batched GEMMs in compact-WY form using
the same ``T^{-1} = diag(1/tau) + strict_upper(V^T V)`` identity as
``ops/linalg.py::_wy_backtransform``.  The remaining half (band ->
tridiagonal bulge chase, reference analog of rocSOLVER's internal
sytrd path) is the round-3 HIP kernel; until it exists this module is
not wired into ``mat_eig_multi`` and carries no env flag.

Reference analog: the ``tcmm_symeig`` replacement contract
(/root/reference/packages/tcmm/src/tcmm_kernel.cu:56-116) -- this is
infrastructure toward the hand-written eigensolver, not a separate
feature.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

__all__ = ["band_reduce_batched", "apply_q_batched"]

Panel = Tuple[int, torch.Tensor, torch.Tensor]  # (r0, V, Tinv)


def _panel_wy(P: torch.Tensor):
    """Batched Householder QR of the (N, M, b) panel in compact-WY
    form.  Returns (V unit-lower-trapezoid (N, M, k), Tinv upper
    (N, k, k), R upper-trapezoid (N, k, b)) with the panel's Q equal
    to I - V T V^T, T = Tinv^{-1}."""
    N, M, b = P.shape
    k = min(M, b)
    a, tau = torch.geqrf(P)
    V = a[:, :, :k].tril(-1)
    V.diagonal(dim1=-2, dim2=-1).fill_(1.0)
    R = a[:, :k, :].triu()
    zr = tau == 0
    if bool(zr.any()):
        # degenerate reflectors (H = I): zero v, tau -> 1 keeps T
        # well-defined and the block exact (K-FAC's rank-deficient
        # sample-covariance factors hit this constantly)
        V = V * (~zr).unsqueeze(1)
        tau = torch.where(zr, torch.ones_like(tau), tau)
    S = torch.bmm(V.mT, V)
    Tinv = S.triu(1) + torch.diag_embed(1.0 / tau)
    return V, Tinv, R


def band_reduce_batched(A: torch.Tensor, b: int = 64
                        ) -> Tuple[torch.Tensor, List[Panel]]:
    """Batched orthogonal reduction of symmetric ``A`` (N, n, n) to
    band width ``b`` (dense storage).  Returns ``(B, panels)`` with
    ``A = Q B Q^T``; ``Q`` stays factored as the panel list consumed
    by :func:`apply_q_batched` (the band eigenvector back-transform).

    Cost is ~(4/3) n^3 flops of k >= b GEMMs plus one ``geqrf`` per
    panel -- n/b panel latencies instead of the n column latencies
    every one-stage tridiagonalization pays.
    """
    if A.dim() != 3 or A.shape[-1] != A.shape[-2]:
        raise ValueError(f"expected (N, n, n) symmetric stack, "
                         f"got {tuple(A.shape)}")
    B = A.clone()
    n = B.shape[-1]
    panels: List[Panel] = []
    for j0 in range(0, n - b - 1, b):
        r0 = j0 + b
        M = n - r0
        if M <= 1:
            break
        V, Tinv, R = _panel_wy(B[:, r0:, j0:j0 + b].contiguous())
        k = V.shape[-1]
        # trailing two-sided update C <- H^T C H as a symmetric
        # rank-2k correction: W = Y T - 1/2 V (T^T S1 T), Y = C V
        C = B[:, r0:, r0:]
        Y = torch.bmm(C, V)
        S1 = torch.bmm(V.mT, Y)
        YT = torch.linalg.solve_triangular(Tinv, Y, upper=True,
                                           left=False)
        U = torch.linalg.solve_triangular(Tinv, S1, upper=True,
                                          left=False)        # S1 T
        U = torch.linalg.solve_triangular(Tinv.mT, U, upper=False)
        W = YT - 0.5 * torch.bmm(V, U)
        C -= torch.bmm(V, W.mT) + torch.bmm(W, V.mT)
        # the panel itself becomes [R; 0] (and its symmetric mirror)
        B[:, r0:, j0:j0 + b] = 0.0
        B[:, r0:r0 + k, j0:j0 + b] = R
        B[:, j0:j0 + b, r0:] = 0.0
        B[:, j0:j0 + b, r0:r0 + k] = R.mT
        panels.append((r0, V, Tinv))
    B = 0.5 * (B + B.mT)
    return B, panels


def apply_q_batched(panels: List[Panel], X: torch.Tensor
                    ) -> torch.Tensor:
    """Compute ``Q @ X`` for the factored ``Q`` from
    :func:`band_reduce_batched` (X: (N, n, m)).  Used to back-transform
    band-matrix eigenvectors to the original basis; three batched GEMMs
    plus one small triangular solve per panel."""
    X = X.clone()
    for r0, V, Tinv in reversed(panels):
        Xr = X[:, r0:, :]
        Y = torch.bmm(V.mT, Xr)
        TY = torch.linalg.solve_triangular(Tinv, Y, upper=True)
        Xr -= torch.bmm(V, TY)
    return X
