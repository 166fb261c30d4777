"""Warm-started blocked Jacobi eigensolver -- the GEMM-rate eigensolve
for K-FAC's repeated factor decompositions.

Why this exists (measured, profiles/PERFORMANCE.md round 2): cold
symmetric eigensolves of the big conv factors are the flagship's whole
budget, and every cold one-stage method is bound by the same
per-column tridiagonalization critical path -- rocSOLVER's batched
syevd sits within ~2x of that floor and the hand-written
persistent-panel sytrd (csrc_solver/sytrd_panel.hip) lands at parity,
not a win.  But K-FAC factors are RUNNING AVERAGES (factor_decay
0.95): between consecutive updates the matrix moves a few percent and
its eigenbasis barely rotates.  A blocked two-sided Jacobi iteration
started from the PREVIOUS basis converges in a few batched-GEMM
passes (MFMA f32 via rocBLAS) -- unlike round 1's perturbative
tracker (first-order, diverged on rank-deficient factors) these are
exact orthogonal iterations with a computable residual, and unlike
rocSOLVER's syevdj the implementation exploits near-diagonality
(measured round 1: syevdj on a near-diagonal input is SLOWER than a
cold syevd).

One iteration (batched over the same-dim factors of a bucket):

  * block-coupling map of B (one elementwise pass over the matrix),
  * per matrix a greedy maximum-weight PERFECT MATCHING of the nb
    blocks into nb//2 disjoint pairs (classical largest-pivot Jacobi,
    uniform across the batch so everything stays one dense batch),
  * one batched eigensolve of the (2bs x 2bs) pair subproblems,
  * three batched GEMM gather/apply/scatter passes (columns, rows,
    accumulated V).

Stop when off(B) <= tol * ||A||_F (fp32-class accuracy by default).
Cold starts and non-converging matrices fall back to the library
tier (the caller handles both), and the caller re-anchors with an
exact library solve every N updates to bound fp32 drift of the
carried basis.  Replaces the reference's cuSOLVER ssyevd on the hot
path (reference: packages/tcmm/src/tcmm_kernel.cu:56-116).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

__all__ = ["block_jacobi_eigh_batched", "offdiag_ratio"]


def offdiag_ratio(B: torch.Tensor,
                  ref_norm: Optional[torch.Tensor] = None) -> torch.Tensor:
    """||offdiag(B)||_F / ref (per batch element)."""
    total = B.pow(2).sum(dim=(-2, -1))
    diag = B.diagonal(dim1=-2, dim2=-1).pow(2).sum(-1)
    off = (total - diag).clamp_min(0)
    ref = total.clamp_min(1e-30) if ref_norm is None else \
        ref_norm.pow(2).clamp_min(1e-30)
    return (off / ref).sqrt()


def _pair_eigh(S: torch.Tensor) -> torch.Tensor:
    """Batched symmetric eigensolve of (k, 2bs, 2bs) subproblems;
    returns rotation matrices (eigenvector columns)."""
    if S.is_cuda:
        from kfac_pytorch_amd.ops import _ext
        if _ext.has_solver():
            solver = _ext.load_solver()
            work = S.contiguous()
            try:
                solver.syevdj_batched_(work, -1)
                # eigenvectors left column-major in the buffer
                return work.mT.contiguous()
            except RuntimeError:
                pass
    _w, Q = torch.linalg.eigh(S)
    return Q


def _greedy_matchings(coup: torch.Tensor):
    """Per-matrix greedy maximum-weight perfect matching of the block
    graph from the (m, nb, nb) coupling map (host side).  Returns a
    long tensor (m, nb//2, 2) of block pairs."""
    m, nb, _ = coup.shape
    npairs = nb // 2
    out = torch.empty(m, npairs, 2, dtype=torch.long)
    for k in range(m):
        c = coup[k]
        order = torch.argsort(c.reshape(-1), descending=True).tolist()
        used = [False] * nb
        pairs = []
        for flat in order:
            p, q = divmod(flat, nb)
            if p >= q or used[p] or used[q]:
                continue
            used[p] = used[q] = True
            pairs.append((p, q))
            if len(pairs) == npairs:
                break
        if len(pairs) < npairs:  # fill from leftovers (zero coupling)
            rest = [b for b in range(nb) if not used[b]]
            while len(pairs) < npairs and len(rest) >= 2:
                pairs.append((rest.pop(0), rest.pop(0)))
        out[k] = torch.tensor(pairs, dtype=torch.long)
    return out


def block_jacobi_eigh_batched(
    A: torch.Tensor,
    V0: Optional[torch.Tensor] = None,
    bs: int = 128,
    tol: float = 5e-6,
    max_iters: int = 10,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    """Eigendecompose a batch of symmetric (m, n, n) matrices.

    Returns (d, V, off, iters): A[k] ~= V[k] diag(d[k]) V[k]^T with
    V[k] columns = eigenvectors (eigh contract, unsorted), ``off`` the
    final per-matrix off-diagonal ratio vs ||A||_F, ``iters`` the
    rotation passes used.  ``V0`` (m, n, n) warm-starts the iteration;
    the caller treats ``off > tol`` as non-convergence (fall back)."""
    m, n, _ = A.shape
    dev = A.device
    nb = (n + bs - 1) // bs
    if nb < 2 or m == 0:
        d, V = torch.linalg.eigh(A)
        return d, V, torch.zeros(m, device=dev), 0
    pad = nb * bs - n
    anorm = A.reshape(m, -1).norm(dim=1).clamp_min(1e-30)
    if pad:
        # decoupled pad block with eigenvalues STRICTLY ABOVE the
        # spectrum (padval > ||A||_F >= lambda_max): in a mixed
        # (data, pad) pair subproblem the ascending eigensolve then
        # keeps data eigenvectors in the first block's coordinates --
        # a below-spectrum pad would sort first and the rotation would
        # permute data coordinates into pad positions
        Ap = A.new_zeros(m, n + pad, n + pad)
        Ap[:, :n, :n] = A
        pidx = torch.arange(n, n + pad, device=dev)
        Ap[:, pidx, pidx] = (1.5 * anorm + 1.0).reshape(m, 1)
        if V0 is not None:
            V0p = A.new_zeros(m, n + pad, n + pad)
            V0p[:, :n, :n] = V0
            V0p[:, pidx, pidx] = 1.0
            V0 = V0p
        A = Ap
    nfull = nb * bs
    npairs = nb // 2
    ncov = npairs * 2 * bs

    if V0 is not None:
        B = torch.bmm(V0.mT, torch.bmm(A, V0))
        B = 0.5 * (B + B.mT)
        V = V0.clone()
    else:
        B = A.clone()
        V = torch.eye(nfull, device=dev, dtype=A.dtype) \
            .expand(m, nfull, nfull).clone()

    diagmask = torch.eye(nb, device=dev, dtype=torch.bool)
    ar = torch.arange(bs, device=dev)
    iters = 0
    off = offdiag_ratio(B, anorm)
    while iters < max_iters and not bool((off < tol).all()):
        # block coupling map (squared-Frobenius of each bs x bs tile)
        sq = B.pow(2).reshape(m, nb, bs, nb, bs).sum(dim=(2, 4))
        sq = sq.masked_fill(diagmask, 0.0)
        match = _greedy_matchings(sq.cpu()).to(dev)  # (m, npairs, 2)
        # column index per pair: (m, npairs, 2bs) and flat (m, ncov)
        cols = (match.unsqueeze(-1) * bs + ar).reshape(m, npairs, 2 * bs)
        flat = cols.reshape(m, ncov)
        # --- subproblems (m*npairs, 2bs, 2bs)
        Brows = torch.gather(
            B, 1, flat.unsqueeze(-1).expand(m, ncov, nfull))
        Brows = Brows.reshape(m * npairs, 2 * bs, nfull)
        S = torch.gather(
            Brows, 2,
            cols.reshape(m * npairs, 2 * bs).unsqueeze(1)
            .expand(m * npairs, 2 * bs, 2 * bs))
        S = 0.5 * (S + S.mT)
        R = _pair_eigh(S)
        # --- column apply: B[:, :, pair] <- B[:, :, pair] @ R
        idx_c = flat.unsqueeze(1).expand(m, nfull, ncov)
        Bc = torch.gather(B, 2, idx_c)
        Bc = Bc.reshape(m, nfull, npairs, 2 * bs).permute(0, 2, 1, 3) \
            .reshape(m * npairs, nfull, 2 * bs)
        Bc = torch.bmm(Bc, R)
        Bc = Bc.reshape(m, npairs, nfull, 2 * bs).permute(0, 2, 1, 3) \
            .reshape(m, nfull, ncov)
        B.scatter_(2, idx_c, Bc)
        # --- row apply: B[:, pair, :] <- R^T @ B[:, pair, :]
        idx_r = flat.unsqueeze(-1).expand(m, ncov, nfull)
        Br = torch.gather(B, 1, idx_r).reshape(m * npairs, 2 * bs, nfull)
        Br = torch.bmm(R.mT, Br)
        B.scatter_(1, idx_r, Br.reshape(m, ncov, nfull))
        # --- accumulate V
        Vc = torch.gather(V, 2, idx_c)
        Vc = Vc.reshape(m, nfull, npairs, 2 * bs).permute(0, 2, 1, 3) \
            .reshape(m * npairs, nfull, 2 * bs)
        Vc = torch.bmm(Vc, R)
        Vc = Vc.reshape(m, npairs, nfull, 2 * bs).permute(0, 2, 1, 3) \
            .reshape(m, nfull, ncov)
        V.scatter_(2, idx_c, Vc)
        iters += 1
        off = offdiag_ratio(B, anorm)
    d = B.diagonal(dim1=-2, dim2=-1)[:, :n].clone()
    V = V[:, :n, :n].contiguous() if pad else V
    return d, V, off, iters
