"""Dense symmetric linear algebra for K-FAC factors.

The reference's hot O(m^3) ops (reference: kfac/utils.py:11-30):
``mat_inv`` (Cholesky inverse) and ``mat_eig`` (symmetric eigensolve),
plus the two preconditioning chains
(reference: kfac/kfac_preconditioner_inv.py:156-161 /
kfac_preconditioner_eigen.py:137-144).

GPU dispatch (all cutoffs measured, profiles/bench_solver.log):

* m <= 64: hand-written batched LDS-Jacobi HIP kernel, one launch for
  a whole rank's small factors;
* everything else: bucketed by dim within ~15% (padded with an
  isolated diagonal block) into strided-batched rocSOLVER
  divide-and-conquer eigensolves / Cholesky inversions driven through
  ``_kfac_solver`` (persistent handles + 4 GB device workspaces, pool
  streams, zero host syncs) -- 3-10x the per-matrix library path;
* singletons overlap on the 8-stream async pool;
* CPU falls back to ``torch.linalg.eigh`` / Cholesky so correctness
  never blocks on a kernel (same ``method`` switch as the reference,
  kfac/utils.py:22-30).

The eigen-precondition scale ``V / (dG dA^T + damping)`` is a fused
elementwise HIP kernel on GPU (batched across same-shape layers).
"""

from __future__ import annotations

import os
from typing import Tuple

import torch

__all__ = [
    "add_diagonal_",
    "mat_inv",
    "mat_inv_multi",
    "mat_eig",
    "mat_eig_multi",
    "eigen_precondition",
    "eigen_precondition_multi",
    "inverse_precondition",
    "check_deferred_info",
]

# ---------------------------------------------------------------- info flags
# rocSOLVER writes per-matrix ``info`` words (non-convergence / non-SPD).
# Checking them synchronously would host-sync the sync-free solve path,
# so every batched/pool call accumulates info into one persistent device
# counter per device; the preconditioner calls ``check_deferred_info()``
# right before issuing the NEXT inverse/eigen update (the queue is
# drained there anyway), raising at most one update late instead of
# silently propagating garbage eigenpairs/inverses (the torch
# eigh/cholesky path this replaces raised immediately).
_INFO_FLAGS: dict = {}


def _defer_info(info: torch.Tensor) -> None:
    flag = _INFO_FLAGS.get(info.device)
    if flag is None:
        flag = torch.zeros((), dtype=torch.int64, device=info.device)
        _INFO_FLAGS[info.device] = flag
    flag.add_(info.ne(0).sum())


def check_deferred_info() -> None:
    """Raise if any rocSOLVER call since the last check reported failure
    (Cholesky of a non-SPD factor / eigensolve non-convergence).
    Host-syncs; call where the device queue is already drained."""
    for dev, flag in _INFO_FLAGS.items():
        bad = int(flag.item())
        if bad:
            flag.zero_()
            raise RuntimeError(
                f"{bad} rocSOLVER factorization(s) on {dev} failed since "
                "the last check (non-SPD factor or eigensolve "
                "non-convergence). A K-FAC update consumed those results; "
                "raise damping or inspect the factors.")


def add_diagonal_(X: torch.Tensor, value) -> torch.Tensor:
    """X += value * I, in place (no full diag-matrix materialization --
    the reference builds a dense diag each call,
    kfac/kfac_preconditioner_inv.py:106-107)."""
    X.diagonal().add_(value)
    return X


def mat_inv(x: torch.Tensor, method: str = "cholesky") -> torch.Tensor:
    """Inverse of a symmetric positive-definite factor
    (reference: kfac/utils.py:11-20)."""
    if method == "cholesky":
        u = torch.linalg.cholesky(x)
        return torch.cholesky_inverse(u).contiguous()
    if method == "inv":
        return torch.linalg.inv(x).contiguous()
    raise NotImplementedError(f"mat_inv method {method!r}")


def mat_eig(x: torch.Tensor, method: str = "auto"
            ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Symmetric eigendecomposition -> (eigenvalues, eigenvectors[m, m]).

    Eigenvectors are returned row-consumable as ``Q`` with columns =
    eigenvectors, i.e. ``x = Q diag(d) Q^T`` (same contract as
    ``torch.linalg.eigh``; reference: kfac/utils.py:22-30).

    ``method``:
      * ``"auto"``  -- HIP Jacobi kernel on GPU when available & m small
                       enough, else ``eigh``.
      * ``"eigh"``  -- torch.linalg.eigh (LAPACK / rocSOLVER).
      * ``"jacobi"``-- force the HIP batched Jacobi kernel (GPU only).
    """
    if method in ("auto", "jacobi") and x.is_cuda:
        from kfac_pytorch_amd.ops import _ext
        if _ext.has_jacobi_eigh(x.shape[-1]) or method == "jacobi":
            return _ext.jacobi_eigh(x)
        method = "eigh"
    if method in ("auto", "eigh"):
        d, Q = torch.linalg.eigh(x)
        return d, Q.contiguous()
    raise NotImplementedError(f"mat_eig method {method!r}")


# dispatch cutoffs, measured on MI355X (profiles/bench_solver.log):
# hand-written LDS-Jacobi wins at m <= 64 (0.33 ms/matrix batched);
# batched divide-and-conquer syevd wins everywhere above (3-10x the
# single-matrix path: its tiny tridiagonalization panel kernels only
# fill the chip when batched).  PAD_RATIO buckets nearby dims into one
# padded batch (pad block = -1 diagonal, sorted out exactly below the
# PSD spectrum): <= 15% dim padding (~50% flops on the smallest
# member) buys another 3x+ of batching.
JAC_DISPATCH_MAX = 64
PAD_RATIO = 1.16
# KFAC_CUSTOM_SYTRD=1 routes buckets at or above KFAC_SYTRD_MIN
# (default 1500, padded dim) through the hand-written persistent-panel
# tridiagonalization (sytrd_panel.hip) + pool-stream rocSOLVER stedc +
# the batched WY back-transform below.  OFF by default: measured on
# MI355X (profiles/PERFORMANCE.md round 2) the custom path reaches
# parity-minus with rocSOLVER's batched syevd (4608 x 3: 247 ms custom
# sytrd vs 244 ms for the library's ENTIRE solve) -- both sit on the
# same per-column tridiagonalization critical path, and the grid
# barriers + cross-workgroup reductions that a persistent kernel needs
# cost about what the library's kernel-launch storm does.  Kept as a
# fully tested opt-in (and the measured study that closes the
# "replace the latrd storm" roadmap item).
SYTRD_DISPATCH_MIN = 1500


def _custom_sytrd_on() -> bool:
    if os.environ.get("KFAC_CUSTOM_SYTRD", "0") != "1":
        return False
    from kfac_pytorch_amd.ops import _ext
    if not _ext.has_solver():
        return False
    return hasattr(_ext.load_solver(), "sytrd_batched_custom_")


def _wy_backtransform(stacked: torch.Tensor, tauT: torch.Tensor,
                      Cstack: torch.Tensor) -> torch.Tensor:
    """Back-transform tridiagonal eigenvectors by the factored Q from
    the custom sytrd: returns the (b, n, n) eigenvector tensor with
    COLUMNS = eigenvectors (the eigh contract).

    ``stacked`` holds the scaled reflectors (row j: unit at j+1, v at
    j+2..; junk below); ``Cstack`` is the raw stedc output buffer
    (eigenvectors column-major).  Q = B_0 B_1 ... B_{P-1} with each
    64-wide block in compact WY form I - V T V^T, where
    T^{-1} = diag(1/tau) + strict_upper(V^T V) -- so each block apply
    is three strided-batched GEMMs plus one small batched triangular
    solve, all MFMA-backed, replacing rocSOLVER's sormtr (which runs
    the same math as small serial larft/larfb kernels).
    """
    b, n, _ = stacked.shape
    res = Cstack.mT.contiguous()  # semantic Z (columns = eigvecs of T)
    dev = stacked.device
    col = torch.arange(n, device=dev)
    # the T^{-1} identity holds at ANY block width, so the apply blocks
    # are wider than the factorization panels: k=256 GEMMs run ~3x the
    # rate of k=64 on rocBLAS
    KB = int(os.environ.get("KFAC_WY_BLOCK", "256"))
    for p in reversed(range(0, n - 1, KB)):
        jb = min(KB, (n - 1) - p)
        tau_p = tauT[:, p:p + jb]
        # V^T as rows, columns < unit zeroed, unit explicit in storage
        mask = (col[p:].unsqueeze(0)
                >= (p + 1 + torch.arange(jb, device=dev)).unsqueeze(1))
        Vt = stacked[:, p:p + jb, p:] * mask
        zr = tau_p == 0
        if bool(zr.any()):
            # degenerate reflectors (H = I): zero v, tau -> 1 keeps
            # T well-defined and the block exact
            Vt = Vt * (~zr).unsqueeze(-1)
            tau_p = torch.where(zr, torch.ones_like(tau_p), tau_p)
        S = torch.bmm(Vt, Vt.mT)
        Tinv = S.triu(1) + torch.diag_embed(1.0 / tau_p)
        Y = torch.bmm(Vt, res[:, p:, :])
        TY = torch.linalg.solve_triangular(Tinv, Y, upper=True)
        res[:, p:, :] -= torch.bmm(Vt.mT, TY)
    return res


def _pad_buckets(dims):
    """Greedy bucketing of sorted-desc (dim, idx) pairs: each bucket's
    members are within PAD_RATIO of the leader and get padded to it."""
    buckets = []
    cur = []
    lead = None
    for d, i in dims:
        if lead is None or d * PAD_RATIO >= lead:
            if lead is None:
                lead = d
            cur.append((d, i))
        else:
            buckets.append((lead, cur))
            lead, cur = d, [(d, i)]
    if cur:
        buckets.append((lead, cur))
    return buckets


def mat_eig_multi(mats, method: str = "auto", need_sorted: bool = True):
    """Eigendecompose a list of symmetric matrices the MI355X way:

    * every Jacobi-eligible matrix (m <= JAC_DISPATCH_MAX) goes into
      ONE hand-written LDS-Jacobi kernel launch;
    * the rest is bucketed by dim within PAD_RATIO (padded with an
      isolated -1 diagonal block) into strided-batched rocSOLVER
      divide-and-conquer eigensolves, one bucket per pool slot;
    * leftover singletons are issued async on a persistent
      handle+stream pool (``syevd_pool_``) and joined once.

    torch.linalg.eigh host-syncs per call, so the per-layer loop the
    reference runs serializes ~106 eigensolves per ResNet-50 step
    (~1.7 s measured); this path removes every host sync.

    ``need_sorted=False`` skips the ascending-eigenvalue reorder --
    K-FAC's eigenvalue clamp and implicit-eigen preconditioner are
    order-independent.  Returns a list of (d, Q) aligned with ``mats``;
    Q may be a non-contiguous transposed view.
    """
    out = [None] * len(mats)
    if not mats:
        return out
    use_gpu = mats[0].is_cuda and method in ("auto", "jacobi")
    if use_gpu:
        from kfac_pytorch_amd.ops import _ext
        jac_idx = [i for i, a in enumerate(mats)
                   if a.shape[-1] <= JAC_DISPATCH_MAX
                   and _ext.has_jacobi_eigh(a.shape[-1])]
        if jac_idx:
            results = _ext.jacobi_eigh_batched(
                [mats[i].contiguous() for i in jac_idx])
            for i, (w, V) in zip(jac_idx, results):
                if need_sorted:
                    w, idx = torch.sort(w)
                    V = V[:, idx].contiguous()
                out[i] = (w, V)

    rest = [i for i in range(len(mats)) if out[i] is None]
    if rest and mats[rest[0]].is_cuda and method != "jacobi" \
            and _solver_ok():
        from kfac_pytorch_amd.ops import _ext
        solver = _ext.load_solver()
        dims = sorted(((int(mats[i].shape[-1]), i) for i in rest),
                      reverse=True)
        singles = []
        slot = 0
        issued_on_pool = False
        device = mats[rest[0]].device
        sytrd_min = int(os.environ.get("KFAC_SYTRD_MIN",
                                       SYTRD_DISPATCH_MIN))
        # info words of slot-issued calls are written on POOL streams:
        # accumulate them only after the join orders the torch stream
        # behind the pool, else the check reads uninitialized memory
        pending_infos = []
        use_custom = _custom_sytrd_on()
        custom_jobs = []  # (members, n4, stacked, tauT, D, Cs)
        for n, members in _pad_buckets(dims):
            if use_custom and n >= sytrd_min:
                # custom tier: persistent-panel sytrd on the torch
                # stream (buckets pipeline: this bucket's stedc on pool
                # slots overlaps the next bucket's sytrd), WY
                # back-transform after the single pool join
                n4 = (n + 3) & ~3  # kernel wants n % 4 == 0
                b = len(members)
                stacked = torch.full((b, n4, n4), 0.0, device=device)
                for k, (m, i) in enumerate(members):
                    pad = n4 - m
                    stacked[k, pad:, pad:] = mats[i]
                    if pad:
                        stacked[k].diagonal()[:pad] = -1.0
                try:
                    E, tauT, status = solver.sytrd_batched_custom_(
                        stacked)
                except RuntimeError:
                    # no resident grid for this (n, b): library tier
                    try:
                        W, _info = solver.syevd_batched_(stacked, slot)
                    except RuntimeError:
                        singles.extend(i for _, i in members)
                        continue
                    pending_infos.append(_info)
                    slot += 1
                    issued_on_pool = True
                    for k, (m, i) in enumerate(members):
                        pad = n4 - m
                        Q = stacked[k].mT
                        out[i] = (W[k, pad:],
                                  Q[pad:, pad:] if pad else Q)
                    continue
                pending_infos.append(status)
                D = stacked.diagonal(dim1=1, dim2=2).contiguous()
                Cs = []
                for k in range(b):
                    C, cinfo = solver.stedc_slot_(D[k], E[k], slot % 8)
                    pending_infos.append(cinfo)
                    slot += 1
                    Cs.append(C)
                issued_on_pool = True
                custom_jobs.append((members, n4, stacked, tauT, D, Cs))
                continue
            if len(members) < 2:
                singles.extend(i for _, i in members)
                continue
            b = len(members)
            stacked = torch.full((b, n, n), 0.0, device=device)
            for k, (m, i) in enumerate(members):
                pad = n - m
                stacked[k, pad:, pad:] = mats[i]
                if pad:
                    # isolated pad block: -1 diagonal sorts strictly
                    # below the PSD factor spectrum, so the pad
                    # eigenpairs are exactly W[:pad] after the
                    # ascending-order solve
                    stacked[k].diagonal()[:pad] = -1.0
            try:
                W, _info = solver.syevd_batched_(stacked, slot)
            except RuntimeError:
                singles.extend(i for _, i in members)
                continue
            pending_infos.append(_info)
            slot += 1
            issued_on_pool = True
            for k, (m, i) in enumerate(members):
                pad = n - m
                # rocSOLVER leaves eigenvectors column-major in the
                # buffer: row-major row i = eigenvector i, so .mT gives
                # the eigh contract (columns = eigenvectors); the real
                # eigenpairs are the top m columns / bottom-right rows
                Q = stacked[k].mT
                out[i] = (W[k, pad:], Q[pad:, pad:] if pad else Q)
        if singles:
            # largest-first keeps the 8 pool streams balanced
            singles.sort(key=lambda i: -mats[i].shape[-1])
            work = [mats[i].clone(
                memory_format=torch.contiguous_format) for i in singles]
            try:
                res = solver.syevd_pool_(work)  # joins ALL pool streams
                _defer_info(res[-1])
                for k, i in enumerate(singles):
                    out[i] = (res[k], work[k].mT)
            except RuntimeError:
                # a matrix larger than the persistent workspace covers
                # (e.g. an un-excluded vocab-sized factor): fall back to
                # torch's own eigh for the whole singles set
                solver.join_pool_()
                for i in singles:
                    out[i] = mat_eig(mats[i], method="eigh")
        elif issued_on_pool:
            solver.join_pool_()
        for _pi in pending_infos:
            _defer_info(_pi)
        # WY back-transform of the custom-sytrd buckets (torch stream,
        # after the pool join so every stedc has finished)
        for members, n4, stacked, tauT, D, Cs in custom_jobs:
            res = _wy_backtransform(stacked, tauT, torch.stack(Cs))
            for k, (m, i) in enumerate(members):
                pad = n4 - m
                out[i] = (D[k, pad:],
                          res[k, pad:, pad:] if pad else res[k])
        if os.environ.get("KFAC_EIG_DEBUG"):
            torch.cuda.synchronize()
            import time as _t
            rep = []
            for n, members in _pad_buckets(dims):
                t0 = _t.perf_counter()
                if len(members) < 2:
                    w = [mats[i].clone() for _, i in members]
                    solver.syevd_pool_(w)
                else:
                    b = len(members)
                    st = torch.full((b, n, n), 0.0, device=device)
                    for k, (m, i) in enumerate(members):
                        st[k, n - m:, n - m:] = mats[i]
                    solver.syevd_batched_(st, -1)
                torch.cuda.synchronize()
                rep.append((n, len(members),
                            round((_t.perf_counter() - t0) * 1e3, 1)))
            print("eig buckets (n, count, ms):", rep, flush=True)
        return out

    for i in rest:
        out[i] = mat_eig(mats[i], method="eigh" if method != "jacobi"
                         else "jacobi")
    return out


def _solver_ok() -> bool:
    from kfac_pytorch_amd.ops import _ext
    return _ext.has_solver()


def mat_inv_multi(mats, damp_diag=None):
    """Cholesky-invert a list of damped SPD GPU matrices with the
    pool-stream rocSOLVER path (potrf+potri overlapped), falling back to
    the serial torch path off-GPU.  ``damp_diag`` is an optional aligned
    list of per-matrix diagonal damping values added BEFORE inversion
    (the pi-damped copy the 'inverse' family makes,
    reference: kfac/kfac_preconditioner_inv.py:109-129).

    Returns a list of inverses (new tensors).
    """
    if not mats:
        return []
    work = []
    for k, a in enumerate(mats):
        w = a.clone(memory_format=torch.contiguous_format)
        if damp_diag is not None:
            w.diagonal().add_(damp_diag[k])
        work.append(w)
    if mats[0].is_cuda and _solver_ok():
        from kfac_pytorch_amd.ops import _ext
        solver = _ext.load_solver()
        out = [None] * len(work)
        dims = sorted(((int(w.shape[-1]), i) for i, w in enumerate(work)),
                      reverse=True)
        singles = []
        slot = 0
        issued = False
        pending = []
        pending_infos = []
        trsm_min = int(os.environ.get("KFAC_TRSM_INV_MIN", "512"))
        for n, members in _pad_buckets(dims):
            if n >= trsm_min:
                # GEMM-rate inverse: L = chol; X = trsm(L, I);
                # A^-1 = X^T X -- measured ~2x rocSOLVER's batched
                # potri at K-FAC factor shapes (the potri tail runs
                # small serial trtri/lauum kernels; trsm+syrk are
                # rocBLAS MFMA work).  Sync-free: cholesky_ex info is
                # deferred like every other solver info word.
                b = len(members)
                stacked = torch.zeros(b, n, n, device=work[0].device)
                for k, (m, i) in enumerate(members):
                    pad = n - m
                    stacked[k, pad:, pad:] = work[i]
                    if pad:
                        stacked[k].diagonal()[:pad] = 1.0
                L, _info = torch.linalg.cholesky_ex(stacked)
                _defer_info(_info)
                eye = torch.eye(n, device=stacked.device) \
                    .expand(b, n, n)
                X = torch.linalg.solve_triangular(L, eye, upper=False)
                inv = torch.bmm(X.mT, X)
                for k, (m, i) in enumerate(members):
                    pad = n - m
                    out[i] = inv[k, pad:, pad:] if pad else inv[k]
                continue
            if len(members) < 2:
                singles.extend(i for _, i in members)
                continue
            b = len(members)
            stacked = torch.zeros(b, n, n, device=work[0].device)
            for k, (m, i) in enumerate(members):
                pad = n - m
                stacked[k, pad:, pad:] = work[i]
                if pad:
                    stacked[k].diagonal()[:pad] = 1.0  # SPD identity pad
            try:
                (_info,) = solver.potri_batched_(stacked, slot)
            except RuntimeError:
                singles.extend(i for _, i in members)
                continue
            pending_infos.append(_info)
            slot += 1
            issued = True
            pending.append((stacked, members))
        if singles:
            sw = [work[i] for i in singles]
            try:
                (_info,) = solver.potri_pool_(sw)  # joins ALL pool streams
                _defer_info(_info)
            except RuntimeError:
                # a matrix beyond the persistent workspace (e.g. an
                # un-excluded vocab-sized factor): degrade to the torch
                # Cholesky path for the singles instead of aborting the
                # whole inverse step (mirrors the eig-path fallback)
                solver.join_pool_()
                for i in singles:
                    out[i] = mat_inv(work[i])
                singles = []
        elif issued:
            solver.join_pool_()
        for _pi in pending_infos:
            _defer_info(_pi)
        # mirror AFTER the join so the torch stream sees finished potri
        for stacked, members in pending:
            n = stacked.shape[-1]
            for k, (m, i) in enumerate(members):
                pad = n - m
                w = stacked[k, pad:, pad:]
                out[i] = w.triu(0) + w.triu(1).mT
        for i in singles:
            w = work[i]
            out[i] = w.triu(0) + w.triu(1).mT
        return out
    return [mat_inv(w) for w in work]


def eigen_precondition(QA: torch.Tensor, dA: torch.Tensor,
                       QG: torch.Tensor, dG: torch.Tensor,
                       grad: torch.Tensor, damping: float) -> torch.Tensor:
    """Implicit-eigen preconditioning of one layer's gradient:

        V1 = QG^T @ grad @ QA
        V2 = V1 / (dG dA^T + damping)
        P  = QG @ V2 @ QA^T

    (reference: kfac/kfac_preconditioner_eigen.py:137-144).
    GEMMs run on MFMA via rocBLAS; the rank-1 denominator scale is fused
    on GPU (one kernel, no dG*dA^T materialization).
    """
    v1 = QG.t() @ grad @ QA
    if grad.is_cuda:
        from kfac_pytorch_amd.ops import _ext
        _ext.eigen_scale_(v1, dG, dA, damping)
        v2 = v1
    else:
        v2 = v1.div_(dG.unsqueeze(1) * dA.unsqueeze(0) + damping)
    return QG @ v2 @ QA.t()


def inverse_precondition(inv_A: torch.Tensor, inv_G: torch.Tensor,
                         grad: torch.Tensor) -> torch.Tensor:
    """P = inv_G @ grad @ inv_A
    (reference: kfac/kfac_preconditioner_inv.py:156-161).
    Accepts the group-stacked 3-D form (batched bmm per group block)."""
    if grad.dim() == 3:
        return torch.bmm(inv_G, torch.bmm(grad, inv_A))
    return inv_G @ grad @ inv_A


def eigen_precondition_grouped(QA: torch.Tensor, dA: torch.Tensor,
                               QG: torch.Tensor, dG: torch.Tensor,
                               grad: torch.Tensor,
                               damping: float) -> torch.Tensor:
    """Implicit-eigen preconditioning of a grouped conv's
    group-stacked gradient: QA/dA/QG/dG are (g, ., .) per-group
    eigenbases, grad is (g, dG, dA); every op is one batched bmm plus
    the fused batched denominator kernel on GPU."""
    v1 = torch.bmm(torch.bmm(QG.mT, grad), QA).contiguous()
    if grad.is_cuda:
        from kfac_pytorch_amd.ops import _ext
        _ext.eigen_scale_batched_(v1, dG.contiguous(), dA.contiguous(),
                                  float(damping))
    else:
        v1 = v1 / (dG.unsqueeze(-1) * dA.unsqueeze(-2) + damping)
    return torch.bmm(torch.bmm(QG, v1), QA.mT)


def eigen_precondition_multi(QAs, dAs, QGs, dGs, grads, damping: float):
    """Implicit-eigen preconditioning of MANY layers' gradients, with
    same-shape layers batched through bmm + one batched denominator
    kernel.  A ResNet-50 repeats (dg, da) factor shapes heavily, so the
    per-layer 4-GEMM loop (~300 small launches) collapses into a few
    bmm groups.  Returns a list of preconditioned grads aligned with the
    inputs.
    """
    out = [None] * len(grads)
    groups = {}
    for i, g in enumerate(grads):
        groups.setdefault((int(g.shape[0]), int(g.shape[1])), []).append(i)
    for (dg, da), idxs in groups.items():
        if len(idxs) == 1 or not grads[idxs[0]].is_cuda:
            for i in idxs:
                out[i] = eigen_precondition(QAs[i], dAs[i], QGs[i], dGs[i],
                                            grads[i], damping)
            continue
        from kfac_pytorch_amd.ops import _ext
        QA = torch.stack([QAs[i] for i in idxs])
        QG = torch.stack([QGs[i] for i in idxs])
        dA = torch.stack([dAs[i] for i in idxs])
        dG = torch.stack([dGs[i] for i in idxs])
        G = torch.stack([grads[i] for i in idxs])
        v1 = torch.bmm(torch.bmm(QG.mT, G), QA).contiguous()
        _ext.eigen_scale_batched_(v1, dG.contiguous(), dA.contiguous(),
                                  float(damping))
        pred = torch.bmm(torch.bmm(QG, v1), QA.mT)
        for k, i in enumerate(idxs):
            out[i] = pred[k]
    return out
