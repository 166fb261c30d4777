"""DP-KFAC with explicit factor inversion ('inverse_dp').

The paper's distributed-preconditioning algorithm (reference:
kfac/kfac_preconditioner_inv_dp.py; DP-KFAC, IEEE TCC 2022): there is
**no factor communication at all** -- hooks save activations/gradients
only on each layer's owner rank, the owner builds factors from its
*local* minibatch, inverts, preconditions the globally-averaged
gradient, and only the preconditioned gradient is broadcast.

MI355X comm layout: the single remaining collective phase (pred
broadcast) is one flat bucket per owner rank on rotating RCCL
communicators.
"""

from __future__ import annotations

import torch

from kfac_pytorch_amd.ops.linalg import inverse_precondition
from kfac_pytorch_amd.preconditioner.inverse import KFACInverse


class KFACInverseDP(KFACInverse):
    """Distributed-preconditioning K-FAC, explicit inverses
    (reference class: kfac/kfac_preconditioner_inv_dp.py:17)."""

    def __init__(self, model, lr=0.1, damping=0.001, fac_update_freq=1,
                 kfac_update_freq=1, kl_clip=0.001, factor_decay=0.95,
                 exclude_vocabulary_size=None, hook_enabled=True,
                 exclude_parts=''):
        super().__init__(model=model, lr=lr, damping=damping,
                         fac_update_freq=fac_update_freq,
                         kfac_update_freq=kfac_update_freq,
                         communicate_inverse_or_not=False,  # forced (ref :49)
                         kl_clip=kl_clip, factor_decay=factor_decay,
                         exclude_vocabulary_size=exclude_vocabulary_size,
                         hook_enabled=hook_enabled,
                         exclude_parts=exclude_parts)
        # schedule eagerly so the owner-gated hooks can fire from step 0
        # (reference :56-57)
        self.schedule_module_ranks()

    # ------------------------------------------------- owner-gated capture
    def _forward_hook_event(self, module, input):
        """Save input only on the owner rank (reference :60-65)."""
        if self._save_input_enabled():
            rank_a, _ = self.module_ranks[module]
            if self.comm.rank() == rank_a:
                self.m_a[module] = input[0].data
                self._overlap_factor(module, "A", input[0].data)

    def _backward_hook_event(self, module, grad_input, grad_output):
        """Save grad-output only on the owner rank (reference :67-72)."""
        if self._save_grad_enabled():
            _, rank_g = self.module_ranks[module]
            if self.comm.rank() == rank_g:
                self.m_g[module] = grad_output[0].data
                self._overlap_factor(module, "G", grad_output[0].data)

    # ---------------------------------------------------------------- state
    def _init_state(self):
        self._alloc_factor_buckets(owner_only=True)
        self._alloc_pred_buckets()

    # -------------------------------------------------------------- factors
    def _compute_factors(self):
        """Owner-local factors only (reference :75-90); layers whose
        fresh factor was already side-computed under backward
        (_overlap_factor) only get the running-average AXPY here."""
        from kfac_pytorch_amd.ops.factors import update_running_avg
        done = self._consume_overlapped()
        rank = self.comm.rank()
        for m in self.modules:
            rank_a, rank_g = self.module_ranks[m]
            if rank == rank_a:
                if (m, "A") in done:
                    update_running_avg(self._ov_fresh[(m, "A")],
                                       self.m_A[m], self.factor_decay)
                else:
                    self.computeA(self.m_a[m], m, out=self.m_A[m],
                                  decay=self.factor_decay)
            if rank == rank_g:
                if (m, "G") in done:
                    update_running_avg(self._ov_fresh[(m, "G")],
                                       self.m_G[m], self.factor_decay)
                else:
                    self.computeG(self.m_g[m], m, batch_averaged=True,
                                  out=self.m_G[m], decay=self.factor_decay)

    def _communicate_factors(self):
        """No factor communication -- the whole point of DP-KFAC
        (reference :93-95)."""

    # ------------------------------------------------------------- inverses
    def _compute_inverse(self):
        """Owner inverts its own locally-built factors (reference
        :98-123), all of them issued as one pool-overlapped potrf+potri
        batch on GPU (mat_inv_multi)."""
        from kfac_pytorch_amd.ops.factors import factor_groups
        from kfac_pytorch_amd.ops.linalg import mat_inv_multi
        rank = self.comm.rank()
        sqrt_damp = self.damping ** 0.5
        mats, damps, dests = [], [], []
        for m in self.modules:
            rank_a, rank_g = self.module_ranks[m]
            if rank != rank_a and rank != rank_g:
                continue
            if rank_a == rank_g and rank == rank_a:
                pi = self._pi_damping(m)
            else:
                pi = None
            gr = factor_groups(m)
            for gi in (range(gr) if gr > 1 else (None,)):
                if pi is None:
                    pig = 1.0
                else:
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

    # ----------------------------------------------------------------- pred
    def _compute_pred(self):
        """Owner preconditions the globally-averaged gradient
        (reference :126-138); non-owner bucket views stay zero as the
        broadcast destination.  Launch-bound phase -> one captured
        hipGraph (base._run_graphed)."""
        assert not self.communicate_inverse_or_not
        rank = self.comm.rank()
        owned = []
        for m in self.modules:
            rank_a, rank_g = self.module_ranks[m]
            assert rank_a == rank_g
            if rank == rank_a:
                owned.append(m)
        if not owned:
            return

        def fn():
            for m in owned:
                grad = self._get_grad(m)
                self.m_precon_grad[m].copy_(inverse_precondition(
                    self.m_inv_A[m], self.m_inv_G[m], grad))

        fp = (tuple(m.weight.grad.data_ptr() for m in owned),
              tuple(m.bias.grad.data_ptr() for m in owned
                    if m.bias is not None),
              tuple(self.m_inv_A[m].data_ptr() for m in owned))
        self._run_graphed("pred", fn, fp)
