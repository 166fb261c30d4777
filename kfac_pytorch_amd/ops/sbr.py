"""Two-stage SBR tridiagonalization, batched end to end
(docs/SBR_STAGE2_NOTES.md; numpy oracle scripts/sbr_ref.py, validated
to machine precision).

Stage 1 (full -> band b): each panel is annihilated with ONE batched
panel factorization (no per-column sequencing -- the measured cost of
every one-stage tridiagonalization, docs/SYTRD_DESIGN.md), and the
two-sided update runs as four batched GEMMs in compact-WY form using
the same ``T^{-1} = diag(1/tau) + strict_upper(V^T V)`` identity as
``ops/linalg.py::_wy_backtransform``.  Measured on MI355X
(profiles/sbr_stage1_deferred.log).

Stage 2 (band -> tridiagonal): TWO validated chase implementations --
the per-element Givens oracle and the reflector-blocked chase the
wavefront HIP kernel runs hop for hop (both fix e1, so implicit-Q
forces the same tridiagonal up to signs).  The torch chases are
correctness references: sequential at Python speed, so this module is
not wired into ``mat_eig_multi`` -- the next-round kernel replaces
exactly the chase inner loop of :func:`sbr_eigh_batched`, which
already passes eigh-parity tests batched.

Reference analog: the ``tcmm_symeig`` replacement contract
(/root/reference/packages/tcmm/src/tcmm_kernel.cu:56-116) -- this is
infrastructure toward the hand-written eigensolver, not a separate
feature.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

__all__ = ["band_reduce_batched", "apply_q_batched",
           "bulge_chase_batched", "bulge_chase_blocked_batched",
           "sbr_eigh_batched"]

Panel = Tuple[int, torch.Tensor, torch.Tensor]  # (r0, V, Tinv)


def _panel_wy_geqrf(P: torch.Tensor):
    """Householder QR of the (N, M, b) panel in compact-WY form via
    batched ``geqrf`` -- the always-correct path (handles rank
    deficiency via tau=0 columns), but latency-bound on ROCm (geqrf
    runs per matrix with internal syncs; measured 187 ms at 4608x3).
    Returns (V unit-lower-trapezoid (N, M, k), Tinv upper (N, k, k),
    R upper-trapezoid (N, k, b)), the panel's Q = I - V T V^T."""
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


def _panel_wy(P: torch.Tensor):
    """Compact-WY panel factorization from batched ops ONLY (the
    MI355X fast path -- no geqrf, no per-matrix library loops):

    1. CholeskyQR with the Gram matrix in fp64 (exact orthonormality
       for any cond(P) < 1e8; the b x b fp64 work is negligible),
    2. Householder reconstruction from the orthonormal Q (Ballard,
       Demmel et al.): unpivoted LU of ``Q1 - S`` (S = -sign(diag Q1))
       gives the unit-lower V1; ``V2 = Q2 U^{-1}``,
    3. ``T^{-1} = strict_upper(V^T V) + diag(V^T V)/2`` -- for ANY V
       this T makes I - V T V^T exactly orthogonal, so the similarity
       transform is exact even where Q itself is only approximate.

    ``Q1 - S`` is diagonally dominant by the sign choice, so batched
    partial-pivoted ``lu_factor`` picks the identity permutation for
    ~99.4% of panels (measured over 2000 random panels); any batch
    with a non-trivial pivot or a failed Cholesky (rank-deficient or
    tail panel) falls back to :func:`_panel_wy_geqrf`.
    """
    N, M, b = P.shape
    if M <= b:
        return _panel_wy_geqrf(P)
    bad = torch.zeros((), dtype=torch.bool, device=P.device)
    out = _panel_wy_fast(P, bad)
    if bool(bad):
        return _panel_wy_geqrf(P)
    return out


def _panel_wy_fast(P: torch.Tensor, bad: torch.Tensor):
    """The branch-free body of :func:`_panel_wy`: every failure mode
    (Cholesky info, non-trivial pivot, non-finite output) accumulates
    into the device-side ``bad`` flag instead of a host sync, so
    callers can run the whole reduction without ever blocking the
    stream and validate ONCE at the end (the same sync-free pattern
    as ops/linalg.py's deferred rocSOLVER info checks)."""
    N, M, b = P.shape
    P64 = P.to(torch.float64)
    G = torch.bmm(P64.mT, P64)
    L1, info = torch.linalg.cholesky_ex(G)
    bad |= (info != 0).any()
    Q = torch.linalg.solve_triangular(L1.mT, P64, upper=True,
                                      left=False)
    # second pass restores orthonormality when cond(P) is large
    G2 = torch.bmm(Q.mT, Q)
    L2, info = torch.linalg.cholesky_ex(G2)
    bad |= (info != 0).any()
    Q = torch.linalg.solve_triangular(L2.mT, Q, upper=True, left=False)
    R = torch.bmm(L2.mT, L1.mT)              # P = Q R
    d = Q.diagonal(dim1=-2, dim2=-1)
    s = -torch.sign(d)
    s = torch.where(s == 0, torch.ones_like(s), s)
    Y1 = Q[:, :b, :] - torch.diag_embed(s)
    lu, piv, luinfo = torch.linalg.lu_factor_ex(Y1)
    ident = torch.arange(1, b + 1, device=piv.device, dtype=piv.dtype)
    bad |= (piv != ident).any() | (luinfo != 0).any()
    U = lu.triu()
    V1 = lu.tril(-1)
    V1.diagonal(dim1=-2, dim2=-1).fill_(1.0)
    V2 = torch.linalg.solve_triangular(U, Q[:, b:, :], upper=True,
                                       left=False)
    V = torch.cat([V1, V2], dim=1).to(P.dtype)
    Gv = torch.bmm(V.mT, V)
    Tinv = Gv.triu(1) + torch.diag_embed(
        Gv.diagonal(dim1=-2, dim2=-1) / 2)
    # H^T P = E S R (S diagonal +-1), so the surviving panel block
    Rp = (s.unsqueeze(-1) * R).to(P.dtype)
    bad |= ~(V.isfinite().all() & Tinv.isfinite().all()
             & Rp.isfinite().all())
    return V, Tinv, Rp


def band_reduce_batched(A: torch.Tensor, b: int = 64,
                        check: str = "deferred"
                        ) -> Tuple[torch.Tensor, List[Panel]]:
    """Batched orthogonal reduction of symmetric ``A`` (N, n, n) to
    band width ``b`` (dense storage).  Returns ``(B, panels)`` with
    ``A = Q B Q^T``; ``Q`` stays factored as the panel list consumed
    by :func:`apply_q_batched` (the band eigenvector back-transform).

    Cost is ~(4/3) n^3 flops of k >= b GEMMs plus one batched panel
    factorization per panel -- n/b panel latencies instead of the n
    column latencies every one-stage tridiagonalization pays.

    ``check`` selects the failure-handling mode:

    * ``"deferred"`` (default): branch-free fast panels with ONE
      device->host validation at the end; on any flagged panel
      (rank-deficient factor, non-trivial pivot, non-finite output)
      the WHOLE reduction reruns in ``"geqrf"`` mode.  This keeps the
      loop free of per-panel GPU syncs -- the measured difference is
      136 ms -> the GEMM-bound floor at 4608x3.
    * ``"eager"``: per-panel host checks with per-panel geqrf
      fallback (mixed-path output; no full rerun).
    * ``"geqrf"``: batched Householder QR for every panel -- the
      always-correct reference path.
    """
    if A.dim() != 3 or A.shape[-1] != A.shape[-2]:
        raise ValueError(f"expected (N, n, n) symmetric stack, "
                         f"got {tuple(A.shape)}")
    if check not in ("deferred", "eager", "geqrf"):
        raise ValueError(f"unknown check mode {check!r}")
    B = A.clone()
    n = B.shape[-1]
    panels: List[Panel] = []
    bad = (torch.zeros((), dtype=torch.bool, device=A.device)
           if check == "deferred" else None)
    for j0 in range(0, n - b - 1, b):
        r0 = j0 + b
        M = n - r0
        if M <= 1:
            break
        P = B[:, r0:, j0:j0 + b].contiguous()
        if check == "geqrf" or M <= b:
            V, Tinv, R = _panel_wy_geqrf(P)
        elif check == "deferred":
            V, Tinv, R = _panel_wy_fast(P, bad)
        else:
            V, Tinv, R = _panel_wy(P)
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
    if bad is not None and bool(bad):
        # rare (rank-deficient / pivoting panel): redo everything on
        # the always-correct path -- results above may hold NaNs
        return band_reduce_batched(A, b, check="geqrf")
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


def bulge_chase_batched(B: torch.Tensor, b: int
                        ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stage 2: batched band(b) -> tridiagonal by Givens bulge chasing
    -- the torch port of the validated numpy reference
    (scripts/sbr_ref.py::bulge_chase, same seats: eliminating
    ``A[k+b, k-1]`` in plane ``(k+b-1, k+b)`` fills ``(k+2b, k+b-1)``).
    Returns ``(T, Q2)`` with ``B = Q2 T Q2^T``.

    Per-matrix divergence (a kill element already zero) is handled
    branch-free: the rotation degenerates to +-identity, so the whole
    batch walks the same structural seat sequence -- exactly the
    property the round-3 wavefront HIP kernel needs.  This is the
    CORRECTNESS reference for that kernel (sequential per-element
    rotations at Python speed, fine at test sizes); it is not wired
    into the production eigensolver tiers.
    """
    A = B.clone()
    N, n, _ = A.shape
    Q2 = torch.eye(n, dtype=A.dtype, device=A.device) \
        .expand(N, n, n).contiguous()

    def rot(p, q, pr, pc, kr, kc):
        # per-matrix Givens in plane (p, q) zeroing A[:, kr, kc]
        # against pivot A[:, pr, pc]; identity where both are zero
        piv = A[:, pr, pc]
        kil = A[:, kr, kc]
        r = torch.hypot(piv, kil)
        safe = r > 0
        c = torch.where(safe, piv / r.clamp_min(1e-300), torch.ones_like(r))
        s = torch.where(safe, kil / r.clamp_min(1e-300), torch.zeros_like(r))
        c_ = c.view(N, 1)
        s_ = s.view(N, 1)
        Rp, Rq = A[:, p, :].clone(), A[:, q, :].clone()
        A[:, p, :] = c_ * Rp + s_ * Rq
        A[:, q, :] = -s_ * Rp + c_ * Rq
        Cp, Cq = A[:, :, p].clone(), A[:, :, q].clone()
        A[:, :, p] = c_ * Cp + s_ * Cq
        A[:, :, q] = -s_ * Cp + c_ * Cq
        Gp, Gq = Q2[:, :, p].clone(), Q2[:, :, q].clone()
        Q2[:, :, p] = c_ * Gp + s_ * Gq
        Q2[:, :, q] = -s_ * Gp + c_ * Gq

    for j in range(n - 2):
        for i in range(min(j + b, n - 1), j + 1, -1):
            rot(i - 1, i, i - 1, j, i, j)
            k = i
            while k + b < n:
                r_ = k + b
                rot(r_ - 1, r_, r_ - 1, k - 1, r_, k - 1)
                k = r_
    A = 0.5 * (A + A.mT)
    return A, Q2


def bulge_chase_blocked_batched(B: torch.Tensor, b: int
                                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stage 2, BLOCKED form: band(b) -> tridiagonal with length-b
    Householder reflectors -- one reflector eliminates column j's band
    below the subdiagonal, then each hop QRs the filled b x b bulge
    block and pushes it b rows down (docs/SBR_STAGE2_NOTES.md).  This
    is the exact per-hop math the wavefront HIP kernel runs (the
    per-element :func:`bulge_chase_batched` is its Givens oracle: both
    fix e1, so by the implicit-Q theorem their tridiagonals agree up
    to signs).  Returns ``(T, Q2)`` with ``B = Q2 T Q2^T``.

    Sequential depth per sweep is ~n/b reflector hops instead of
    ~n per-element rotations -- the property that makes one grid
    barrier per macro-step feasible on the GPU.
    """
    A = B.clone()
    N, n, _ = A.shape
    Q2 = torch.eye(n, dtype=A.dtype, device=A.device) \
        .expand(N, n, n).contiguous()
    for j in range(n - 2):
        for _ in _sweep_hops(A, Q2, j, b):
            pass
    A = 0.5 * (A + A.mT)
    return A, Q2


def _apply_chase_block(A, Q2, rs, re, cs, ce):
    """QR the (rs:re, cs:ce) block in-place (rows rs:re mix) and apply
    the block reflector two-sided + into Q2."""
    P = A[:, rs:re, cs:ce].contiguous()
    M, k = P.shape[1], min(P.shape[1], P.shape[2])
    if M < 2:
        return
    V, Tinv, R = _panel_wy_geqrf(P)
    # rows: A[S,:] <- H^T A[S,:]
    W = torch.bmm(V.mT, A[:, rs:re, :])
    TtW = torch.linalg.solve_triangular(Tinv.mT, W, upper=False)
    A[:, rs:re, :] -= torch.bmm(V, TtW)
    # cols: A[:,S] <- A[:,S] H   (and the same for Q2)
    for Mt in (A, Q2):
        Y = torch.bmm(Mt[:, :, rs:re], V)
        YT = torch.linalg.solve_triangular(Tinv, Y, upper=True,
                                           left=False)
        Mt[:, :, rs:re] -= torch.bmm(YT, V.mT)
    # exact zeros on the annihilated block (and its mirror): the top k
    # rows hold H^T P = R
    A[:, rs:re, cs:ce] = 0.0
    A[:, rs:rs + k, cs:ce] = R
    A[:, cs:ce, rs:re] = 0.0
    A[:, cs:ce, rs:rs + k] = R.mT


def _sweep_hops(A, Q2, j, b):
    """Generator over sweep ``j``'s reflector hops (mutates A/Q2 in
    place, yields after each hop).  Hop t's row window is
    ``(j + t*b, min(j + (t+1)*b, n)]`` -- windows of sweeps whose hop
    indices differ by >= 2 are disjoint, which is the wavefront
    pipelining precondition the HIP kernel relies on
    (tests/test_sbr_stage1.py::test_wavefront_interleaving_commutes
    validates it by driving these generators interleaved)."""
    n = A.shape[-1]
    s, e = j + 1, min(j + 1 + b, n)
    if e - s >= 2:
        _apply_chase_block(A, Q2, s, e, j, j + 1)
    yield
    while e < n:
        ns, ne = e, min(e + b, n)
        _apply_chase_block(A, Q2, ns, ne, s, e)
        s, e = ns, ne
        yield


def sbr_eigh_batched(A: torch.Tensor, b: int = 64,
                     chase: str = "blocked"
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Complete two-stage SBR symmetric eigensolve, batched:
    full -> band(b) -> tridiagonal -> eigh, with both back-transforms.
    Returns ``(eigenvalues (N, n), eigenvectors (N, n, n))`` in the
    ``torch.linalg.eigh`` convention (ascending, columns).

    End-to-end correctness pipeline for the round-3 kernel: stage 1
    runs at batched-GEMM rate (:func:`band_reduce_batched`), stage 2
    is the reflector-blocked chase (``chase="blocked"``, the form the
    HIP kernel implements hop for hop) or the per-element Givens
    oracle (``chase="givens"``); the tridiagonal solve stands in for
    the existing stedc binding.  Tested against ``torch.linalg.eigh``.
    """
    Bb, panels = band_reduce_batched(A, b)
    chase_fn = {"blocked": bulge_chase_blocked_batched,
                "givens": bulge_chase_batched}[chase]
    T, Q2 = chase_fn(Bb, b)
    d, Z = torch.linalg.eigh(T)
    vecs = apply_q_batched(panels, torch.bmm(Q2, Z))
    return d, vecs
