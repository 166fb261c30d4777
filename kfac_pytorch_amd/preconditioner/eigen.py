"""MPD K-FAC with implicit eigen-decomposition ('eigen').

SC-20-style (reference: kfac/kfac_preconditioner_eigen.py): factors are
allreduce-averaged; each layer's A and G are eigendecomposed by owner
ranks (optionally factor-wise: A and G of one layer on *different*
ranks when world > #layers); eigenbases (QA, dA, QG, dG) are broadcast;
every rank then computes the implicit-eigen preconditioned gradient.

MI355X comm layout: one flat factor allreduce; eigenbases packed into
one flat bucket per owner rank and broadcast concurrently on rotating
RCCL communicators.
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from kfac_pytorch_amd.ops.factors import factor_dims
from kfac_pytorch_amd.ops.linalg import (eigen_precondition_multi,
                                         mat_eig, mat_eig_multi)
from kfac_pytorch_amd.preconditioner.inverse import KFACInverse


class EigenComputeMixin:
    """Shared owner-side eigendecomposition, MI355X-scheduled
    (dispatch measured in profiles/bench_solver.log):

    * small factors (m <= 64) batch into ONE hand-written LDS-Jacobi
      kernel launch;
    * everything else buckets by dim (within ~15%, padded with an
      isolated -1 diagonal block) into strided-batched rocSOLVER
      divide-and-conquer eigensolves -- 3-10x the single-matrix path,
      whose tiny tridiagonalization panels only fill the chip when
      batched; leftover singletons overlap on a persistent async
      stream pool (no host syncs -- torch.linalg.eigh would host-sync
      per matrix and serialize the whole set);
    * CPU falls back to plain eigh.

    Replaces the reference's serial per-layer eigh loop
    (reference: kfac_preconditioner_eigen.py:98-119).

    ``KFAC_WARM_EIG=1`` routes factors >= KFAC_WARM_MIN (default 512)
    through the warm-started blocked Jacobi tier (ops/block_jacobi.py)
    carrying each factor's eigenbasis across updates: in steady-state
    training (slow factor drift) a couple of batched-GEMM passes
    replace the cold eigensolve.  The dispatch gates on the MEASURED
    residual -- a factor whose basis genuinely rotated falls back to
    the library tier that update (and re-anchors exactly every
    KFAC_WARM_REANCHOR updates to bound fp32 drift), so accuracy never
    depends on the steadiness assumption.  Off by default: at
    fac/kfac_update_freq=1 with fresh sampling noise every step the
    warm tier only breaks even (measured; see
    profiles/PERFORMANCE.md round 2).
    """

    def _eigendecompose_owned(self):
        import os
        from kfac_pytorch_amd.ops.factors import factor_groups
        rank = self.comm.rank()
        work = []  # (module, kind, group-index-or-None)
        for m in self.modules:
            rank_a, rank_g = self.module_ranks[m]
            gr = factor_groups(m)
            gis = range(gr) if gr > 1 else (None,)
            for gi in gis:
                if rank == rank_a:
                    work.append((m, "A", gi))
                if rank == rank_g:
                    work.append((m, "G", gi))
        if not work:
            return
        mats = []
        for mod, kind, gi in work:
            t = self.m_A[mod] if kind == "A" else self.m_G[mod]
            mats.append(t if gi is None else t[gi])
        if (os.environ.get("KFAC_WARM_EIG", "0") == "1"
                and mats[0].is_cuda):
            results = self._warm_eig(work, mats)
        else:
            results = mat_eig_multi(mats, need_sorted=False)

        for (mod, kind, gi), (d, Q) in zip(work, results):
            clamped = d * (d > self.eps)
            tq = self.m_QA[mod] if kind == "A" else self.m_QG[mod]
            td = self.m_dA[mod] if kind == "A" else self.m_dG[mod]
            if gi is None:
                tq.copy_(Q)
                td.copy_(clamped)
            else:
                tq[gi].copy_(Q)
                td[gi].copy_(clamped)

    def _warm_eig(self, work, mats):
        """Warm-tier dispatch: carried-basis blocked Jacobi for the big
        factors, measured-residual gated, library tier for the rest."""
        import os
        import torch as _torch
        from kfac_pytorch_amd.ops.block_jacobi import \
            block_jacobi_eigh_batched
        warm_min = int(os.environ.get("KFAC_WARM_MIN", "512"))
        reanchor = int(os.environ.get("KFAC_WARM_REANCHOR", "50"))
        tol = float(os.environ.get("KFAC_WARM_TOL", "5e-6"))
        max_iters = int(os.environ.get("KFAC_WARM_MAX_ITERS", "12"))
        if not hasattr(self, "_warm_state"):
            self._warm_state = {}  # (id(mod), kind, gi) -> [V, age]
        results = [None] * len(mats)
        groups = {}
        for i, (mod, kind, gi) in enumerate(work):
            key = (id(mod), kind, gi)
            st = self._warm_state.get(key)
            n = int(mats[i].shape[-1])
            # re-anchor exactly (library solve) on a staggered schedule
            if (n >= warm_min and st is not None
                    and st[1] % reanchor != 0):
                groups.setdefault(n, []).append((i, key))
        for n, items in groups.items():
            A = _torch.stack([mats[i] for i, _ in items])
            V0 = _torch.stack([self._warm_state[k][0] for _, k in items])
            d, V, off, _iters = block_jacobi_eigh_batched(
                A, V0=V0, tol=tol, max_iters=max_iters)
            offh = off.cpu()
            if hasattr(self, "phase_times"):  # KFAC_PHASE_TIMING stats
                self.phase_times["warm_eig_iters"] = \
                    self.phase_times.get("warm_eig_iters", 0) + _iters
            for j, (i, k) in enumerate(items):
                if float(offh[j]) < tol:
                    results[i] = (d[j], V[j])
                    self._warm_state[k][0] = V[j].contiguous()
                    self._warm_state[k][1] += 1
                    if hasattr(self, "phase_times"):
                        self.phase_times["warm_eig_hits"] = \
                            self.phase_times.get("warm_eig_hits", 0) + 1
                else:
                    # basis genuinely rotated: cold-solve below
                    self._warm_state.pop(k, None)
                    if hasattr(self, "phase_times"):
                        self.phase_times["warm_eig_miss"] = \
                            self.phase_times.get("warm_eig_miss", 0) + 1
        rest = [i for i in range(len(mats)) if results[i] is None]
        if rest:
            lib = mat_eig_multi([mats[i] for i in rest],
                                need_sorted=False)
            for i, r in zip(rest, lib):
                results[i] = r
                n = int(mats[i].shape[-1])
                if n >= warm_min:
                    key = (id(work[i][0]), work[i][1], work[i][2])
                    st = self._warm_state.get(key)
                    # stagger each key's re-anchor phase so no single
                    # step pays every exact solve at once
                    age = st[1] + 1 if st else \
                        1 + (len(self._warm_state) * 7) % reanchor
                    self._warm_state[key] = [r[1].contiguous(), age]
        return results


class KFACEigen(EigenComputeMixin, KFACInverse):
    """Model-parallel distributed K-FAC, implicit eigen preconditioning
    (reference class: kfac/kfac_preconditioner_eigen.py:18)."""

    def __init__(self, model, lr=0.1, damping=0.001, fac_update_freq=1,
                 kfac_update_freq=1, distribute_layer_factors=None,
                 kl_clip=0.001, factor_decay=0.95,
                 exclude_vocabulary_size=None, hook_enabled=True,
                 exclude_parts=''):
        super().__init__(model=model, lr=lr, damping=damping,
                         fac_update_freq=fac_update_freq,
                         kfac_update_freq=kfac_update_freq,
                         communicate_inverse_or_not=True,  # forced (ref :52)
                         kl_clip=kl_clip, factor_decay=factor_decay,
                         exclude_vocabulary_size=exclude_vocabulary_size,
                         hook_enabled=hook_enabled,
                         exclude_parts=exclude_parts)
        self.m_QA: Dict[nn.Module, torch.Tensor] = {}
        self.m_QG: Dict[nn.Module, torch.Tensor] = {}
        self.m_dA: Dict[nn.Module, torch.Tensor] = {}
        self.m_dG: Dict[nn.Module, torch.Tensor] = {}
        self._distribute_layer_factors = distribute_layer_factors

    # ------------------------------------------------------------- schedule
    def schedule_module_ranks(self):
        """Round-robin; factor-wise (rank_g = rank_a + 1) when the world
        outnumbers the layers (reference :66-71,75-94)."""
        if self._distribute_layer_factors is None:
            factor_wise = self.comm.size() > len(self.modules)
        else:
            factor_wise = self._distribute_layer_factors
        return self._round_robin_ranks(factor_wise=factor_wise)

    # ---------------------------------------------------------------- state
    def _init_state(self):
        self._alloc_factor_buckets(owner_only=False)
        specs = []
        for i, m in enumerate(self.modules):
            rank_a, rank_g = self.module_ranks[m]
            da, dg = factor_dims(m)
            specs.append((f"QA{i}", self._block_shape(m, da), rank_a))
            specs.append((f"dA{i}", self._block_shape(m, da, vec=True),
                          rank_a))
            specs.append((f"QG{i}", self._block_shape(m, dg), rank_g))
            specs.append((f"dG{i}", self._block_shape(m, dg, vec=True),
                          rank_g))
        self.eig_buckets = self._alloc_owner_buckets(specs)
        for i, m in enumerate(self.modules):
            self.m_QA[m] = self._owner_view(self.eig_buckets, f"QA{i}")
            self.m_dA[m] = self._owner_view(self.eig_buckets, f"dA{i}")
            self.m_QG[m] = self._owner_view(self.eig_buckets, f"QG{i}")
            self.m_dG[m] = self._owner_view(self.eig_buckets, f"dG{i}")

    # ------------------------------------------------------------- inverses
    def _use_multibcast(self) -> bool:
        """KFAC_NATIVE_MULTIBCAST=1: the reference's fused tcmm
        research path (per-factor eigensolve on the multi_bcast owner,
        overlapped rotating-comm broadcasts of packed outputs) instead
        of the default batched-solve + flat-bucket broadcast.  Needs
        the native communicator (world > 1, GPU, KFAC_NATIVE_COMM=1)."""
        import os
        return (os.environ.get("KFAC_NATIVE_MULTIBCAST") == "1"
                and self._native_comm() is not None)

    def _compute_inverse(self):
        """Owner ranks eigendecompose their factors (batched); eigenvalues
        clamped at eps (reference :98-119)."""
        if self._use_multibcast():
            return  # fused into _communicate_inverse
        self._eigendecompose_owned()

    def _communicate_inverse(self):
        if self._use_multibcast():
            self._fused_eigen_multibcast()
            return
        self._broadcast_owner_buckets(self.eig_buckets)

    def _fused_eigen_multibcast(self):
        """Fused compute+broadcast of every factor's eigendecomposition
        (reference research path: communicator.cpp:75-117,
        scripts/bench_ops.py:111-146).  The multi_bcast round-robin
        owner schedule replaces module_ranks for THIS phase; outputs
        land identically on every rank, which is all the MPD eigen
        pred phase needs."""
        from kfac_pytorch_amd.parallel.native import \
            fused_eigen_multibcast
        from kfac_pytorch_amd.ops.factors import factor_groups
        nat = self._native_comm()
        facs, outs, metas = [], [], []
        for m in self.modules:
            gr = factor_groups(m)
            for kind, fac in (("A", self.m_A[m]), ("G", self.m_G[m])):
                for gi in (range(gr) if gr > 1 else (None,)):
                    f2 = fac if gi is None else fac[gi]
                    n = f2.shape[0]
                    facs.append(f2.contiguous())
                    outs.append(f2.new_empty(n, n + 1))
                    metas.append((m, kind, gi, n))
        fused_eigen_multibcast(nat, facs, outs)
        for (m, kind, gi, n), out in zip(metas, outs):
            d = out[:, n]
            clamped = d * (d > self.eps)
            tq = self.m_QA[m] if kind == "A" else self.m_QG[m]
            td = self.m_dA[m] if kind == "A" else self.m_dG[m]
            if gi is None:
                tq.copy_(out[:, :n])
                td.copy_(clamped)
            else:
                tq[gi].copy_(out[:, :n])
                td[gi].copy_(clamped)

    # ----------------------------------------------------------------- pred
    def _compute_pred(self):
        """Implicit-eigen preconditioning on every rank (reference
        :137-144), same-shape layers batched, replayed as one hipGraph
        (launch-bound phase; see base._run_graphed)."""
        from kfac_pytorch_amd.ops.factors import factor_groups
        from kfac_pytorch_amd.ops.linalg import eigen_precondition_grouped
        mods = [m for m in self.modules if factor_groups(m) == 1]
        gmods = [m for m in self.modules if factor_groups(m) > 1]

        def fn():
            preds = eigen_precondition_multi(
                [self.m_QA[m] for m in mods],
                [self.m_dA[m] for m in mods],
                [self.m_QG[m] for m in mods],
                [self.m_dG[m] for m in mods],
                [self._get_grad(m) for m in mods], self.damping)
            for m in gmods:
                preds.append(eigen_precondition_grouped(
                    self.m_QA[m], self.m_dA[m], self.m_QG[m],
                    self.m_dG[m], self._get_grad(m), self.damping))
            for m, p in zip(mods + gmods, preds):
                if m in self.m_precon_grad and \
                        self.m_precon_grad[m].shape == p.shape:
                    self.m_precon_grad[m].copy_(p)
                else:
                    self.m_precon_grad[m] = p

        if any(m not in self.m_precon_grad for m in mods + gmods):
            # first step allocates the output buffers eagerly so the
            # captured graph only ever writes into static storage
            fn()
            return
        allm = mods + gmods
        fp = (float(self.damping),
              tuple(m.weight.grad.data_ptr() for m in allm),
              tuple(m.bias.grad.data_ptr() for m in allm
                    if m.bias is not None),
              tuple(self.m_QA[m].data_ptr() for m in allm))
        self._run_graphed("pred", fn, fp)
