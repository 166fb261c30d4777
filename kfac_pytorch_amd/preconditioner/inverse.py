"""MPD K-FAC with explicit factor inversion ('inverse').

CVPR-19-style model-parallel distributed preconditioning
(reference: kfac/kfac_preconditioner_inv.py): factors are
allreduce-averaged, each layer's inverse is computed by one owner rank
(round-robin, rank_a == rank_g), and either the inverses or the
owner-computed preconditioned gradients are broadcast back
(``communicate_inverse_or_not``, default False -> broadcast pred).

MI355X comm layout: the factor allreduce is ONE collective over a flat
buffer; inverse / pred broadcasts are one flat bucket per owner rank,
issued concurrently on rotating RCCL communicators.
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from kfac_pytorch_amd.ops.factors import ComputeA, ComputeG, factor_dims
from kfac_pytorch_amd.ops.linalg import (inverse_precondition,
                                         mat_inv)
from kfac_pytorch_amd.preconditioner.base import KFACBase


class KFACInverse(KFACBase):
    """Model-parallel distributed K-FAC, explicit inverses
    (reference class: kfac/kfac_preconditioner_inv.py:17)."""

    def __init__(self, model, lr=0.1, damping=0.001, fac_update_freq=1,
                 kfac_update_freq=1, communicate_inverse_or_not=False,
                 kl_clip=0.001, factor_decay=0.95,
                 exclude_vocabulary_size=None, hook_enabled=True,
                 exclude_parts=''):
        super().__init__(model=model, lr=lr, damping=damping,
                         fac_update_freq=fac_update_freq,
                         kfac_update_freq=kfac_update_freq,
                         communicate_inverse_or_not=communicate_inverse_or_not,
                         kl_clip=kl_clip, factor_decay=factor_decay,
                         exclude_vocabulary_size=exclude_vocabulary_size,
                         hook_enabled=hook_enabled,
                         exclude_parts=exclude_parts)
        self.computeA = ComputeA()
        self.computeG = ComputeG()
        self.m_inv_A: Dict[nn.Module, torch.Tensor] = {}
        self.m_inv_G: Dict[nn.Module, torch.Tensor] = {}

    # ------------------------------------------------------------- schedule
    def schedule_module_ranks(self):
        return self._round_robin_ranks(factor_wise=False)

    # ---------------------------------------------------------------- state
    def _init_state(self):
        self._alloc_factor_buckets(owner_only=False)
        if self.communicate_inverse_or_not:
            self._alloc_inverse_buckets()
        else:
            self._alloc_pred_buckets()

    def _alloc_inverse_buckets(self):
        specs = []
        for i, m in enumerate(self.modules):
            rank_a, rank_g = self.module_ranks[m]
            da, dg = factor_dims(m)
            specs.append((f"invA{i}", self._block_shape(m, da), rank_a))
            specs.append((f"invG{i}", self._block_shape(m, dg), rank_g))
        self.inv_buckets = self._alloc_owner_buckets(specs)
        for i, m in enumerate(self.modules):
            self.m_inv_A[m] = self._owner_view(self.inv_buckets, f"invA{i}")
            self.m_inv_G[m] = self._owner_view(self.inv_buckets, f"invG{i}")

    def _alloc_pred_buckets(self):
        from kfac_pytorch_amd.ops.factors import factor_groups
        specs = []
        for i, m in enumerate(self.modules):
            rank_a, _ = self.module_ranks[m]
            da, dg = factor_dims(m)
            gr = factor_groups(m)
            shape = (gr, dg, da) if gr > 1 else (dg, da)
            specs.append((f"pred{i}", shape, rank_a))
        self.pred_buckets = self._alloc_owner_buckets(specs)
        for i, m in enumerate(self.modules):
            self.m_precon_grad[m] = self._owner_view(
                self.pred_buckets, f"pred{i}")
        # inverses live only on the owner (never broadcast in this mode)
        for m in self.modules:
            da, dg = factor_dims(m)
            dev = self._state_device()
            rank_a, rank_g = self.module_ranks[m]
            if self.comm.rank() == rank_a:
                self.m_inv_A[m] = torch.zeros(
                    self._block_shape(m, da), device=dev)
            if self.comm.rank() == rank_g:
                self.m_inv_G[m] = torch.zeros(
                    self._block_shape(m, dg), device=dev)

    # -------------------------------------------------------------- factors
    def _compute_factors(self):
        """Local factors + running average, every rank, every layer
        (reference :80-91); the running-average update is fused into the
        factor kernel via out=/decay= -- except for layers whose fresh
        factor was already side-computed under backward
        (_overlap_factor), which only need the AXPY here."""
        from kfac_pytorch_amd.ops.factors import update_running_avg
        done = self._consume_overlapped()
        for m in self.modules:
            if (m, "A") in done:
                update_running_avg(self._ov_fresh[(m, "A")],
                                   self.m_A[m], self.factor_decay)
            else:
                self.computeA(self.m_a[m], m, out=self.m_A[m],
                              decay=self.factor_decay)
            if (m, "G") in done:
                update_running_avg(self._ov_fresh[(m, "G")],
                                   self.m_G[m], self.factor_decay)
            else:
                self.computeG(self.m_g[m], m, batch_averaged=True,
                              out=self.m_G[m], decay=self.factor_decay)

    def _communicate_factors(self):
        """ONE flat allreduce-average for all layers' A and G
        (vs per-layer async bursts, reference :94-103)."""
        self._allreduce_bucket_avg(self.factor_bucket)

    # ------------------------------------------------------------- inverses
    def _pi_damping(self, m) -> torch.Tensor:
        """pi = sqrt((trA/dimA)/(trG/dimG)) (reference :121); for
        grouped convs one pi per block, shape (g,)."""
        A, G = self.m_A[m], self.m_G[m]
        trA = A.diagonal(dim1=-2, dim2=-1).sum(-1) / A.shape[-1]
        trG = G.diagonal(dim1=-2, dim2=-1).sum(-1) / G.shape[-1]
        return torch.sqrt(trA / trG)

    def _compute_inverse(self):
        """Owner rank inverts its layers' damped factors (reference
        :109-129). The damped copy is temporary; m_A/m_G stay undamped
        running averages.  All of a rank's inversions are issued as one
        pool-overlapped potrf+potri batch on GPU (mat_inv_multi)."""
        from kfac_pytorch_amd.ops.linalg import mat_inv_multi
        rank = self.comm.rank()
        sqrt_damp = self.damping ** 0.5
        from kfac_pytorch_amd.ops.factors import factor_groups
        mats, damps, dests = [], [], []
        for m in self.modules:
            rank_a, rank_g = self.module_ranks[m]
            if rank != rank_a and rank != rank_g:
                continue
            pi = self._pi_damping(m)
            gr = factor_groups(m)
            for gi in (range(gr) if gr > 1 else (None,)):
                pig = pi if gi is None else pi[gi]
                if rank == rank_a:
                    mats.append(self.m_A[m] if gi is None
                                else self.m_A[m][gi])
                    damps.append(sqrt_damp * pig)
                    dests.append(self.m_inv_A[m] if gi is None
                                 else self.m_inv_A[m][gi])
                if rank == rank_g:
                    mats.append(self.m_G[m] if gi is None
                                else self.m_G[m][gi])
                    damps.append(sqrt_damp / pig)
                    dests.append(self.m_inv_G[m] if gi is None
                                 else self.m_inv_G[m][gi])
        for inv, dst in zip(mat_inv_multi(mats, damp_diag=damps), dests):
            dst.copy_(inv)

    def _communicate_inverse(self):
        self._broadcast_owner_buckets(self.inv_buckets)

    # ----------------------------------------------------------------- pred
    def _compute_pred(self):
        """Per-layer inv_G @ grad @ inv_A; launch-bound, so the whole
        phase replays as one captured hipGraph (base._run_graphed)."""
        if self.communicate_inverse_or_not:
            # every rank preconditions every layer with broadcast inverses
            mods = self.modules

            def fn():
                for m in mods:
                    grad = self._get_grad(m)
                    p = inverse_precondition(
                        self.m_inv_A[m], self.m_inv_G[m], grad)
                    if m in self.m_precon_grad and \
                            self.m_precon_grad[m].shape == p.shape:
                        self.m_precon_grad[m].copy_(p)
                    else:
                        self.m_precon_grad[m] = p

            if any(m not in self.m_precon_grad for m in mods):
                fn()  # first step allocates static output storage
                return
        else:
            # owner-only pred, broadcast after (reference default :41)
            rank = self.comm.rank()
            mods = [m for m in self.modules
                    if rank == self.module_ranks[m][0]]
            if not mods:
                return

            def fn():
                for m in mods:
                    grad = self._get_grad(m)
                    self.m_precon_grad[m].copy_(inverse_precondition(
                        self.m_inv_A[m], self.m_inv_G[m], grad))

        fp = (tuple(m.weight.grad.data_ptr() for m in mods),
              tuple(m.bias.grad.data_ptr() for m in mods
                    if m.bias is not None),
              tuple(self.m_inv_A[m].data_ptr() for m in mods))
        self._run_graphed("pred", fn, fp)

    def _communicate_pred(self):
        self._broadcast_owner_buckets(self.pred_buckets)
