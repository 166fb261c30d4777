"""Distributed K-FAC preconditioner base class.

Keeps the reference's public contract
(reference: kfac/kfac_preconditioner_base.py:13-231):

* ``optim.Optimizer`` subclass so ``LambdaLR`` / ``KFACParamScheduler``
  drive ``lr``/``damping``/``*_update_freq`` through ``param_groups``.
* forward-pre / full-backward hooks on every ``Linear``/``Conv2d`` save
  activations and output-gradients by reference (zero-copy until
  ``step()`` consumes them).
* the 4-phase ``step()`` template: compute/communicate factors ->
  compute/communicate inverses -> compute/communicate preconditioned
  gradients -> in-place gradient update with kl-clip; ``exclude_parts``
  ablation flags skip individual phases for time breakdowns.

MI355X-first changes (not in the reference):

* All per-layer state lives in :class:`FlatBucket` views so each
  communication phase is one (or per-owner one) large collective instead
  of a burst of per-layer <=1 MB messages -- the xGMI links are
  point-to-point and latency-bound for small tensors
  (SURVEY.md S5 'Distributed communication backend').
* Owner-rooted broadcasts are issued concurrently on rotating duplicate
  process groups (separate RCCL comms/streams per group).
* kl-clip is computed without any host-device sync (the reference calls
  ``.item()`` per layer, kfac/kfac_preconditioner_inv.py:198-200).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn
import torch.optim as optim

import kfac_pytorch_amd.parallel.comm as comm_mod
from kfac_pytorch_amd.parallel.comm import FlatBucket

logger = logging.getLogger(__name__)

SUPPORTED_MODULES = ("Linear", "Conv2d")


class KFACBase(optim.Optimizer):
    """Base distributed K-FAC gradient preconditioner.

    Args mirror the reference (kfac/kfac_preconditioner_base.py:53-77):
      model, lr, damping, fac_update_freq, kfac_update_freq,
      communicate_inverse_or_not, kl_clip, factor_decay,
      exclude_vocabulary_size, hook_enabled, exclude_parts.
    """

    def __init__(self,
                 model: nn.Module,
                 lr: float = 0.1,
                 damping: float = 0.001,
                 fac_update_freq: int = 1,
                 kfac_update_freq: int = 1,
                 communicate_inverse_or_not: bool = True,
                 kl_clip: float = 0.001,
                 factor_decay: float = 0.95,
                 exclude_vocabulary_size: Optional[int] = None,
                 hook_enabled: bool = True,
                 exclude_parts: str = ''):

        defaults = dict(lr=lr, damping=damping,
                        fac_update_freq=fac_update_freq,
                        kfac_update_freq=kfac_update_freq)
        super().__init__(model.parameters(), defaults)

        self.lr = lr
        self.damping = damping
        self.fac_update_freq = fac_update_freq
        self.kfac_update_freq = kfac_update_freq
        self.communicate_inverse_or_not = communicate_inverse_or_not
        self.kl_clip = kl_clip if (kl_clip is not None and kl_clip > 0) else None
        self.factor_decay = factor_decay
        self.exclude_vocabulary_size = exclude_vocabulary_size
        self.hook_enabled = hook_enabled

        self.exclude_communicate_inverse = 'CommunicateInverse' in exclude_parts
        self.exclude_compute_inverse = 'ComputeInverse' in exclude_parts
        self.exclude_communicate_factor = 'CommunicateFactor' in exclude_parts
        self.exclude_compute_factor = 'ComputeFactor' in exclude_parts

        self.modules: List[nn.Module] = []
        self.module_names: List[str] = []
        self._hook_handles = []
        self._register_module_hooks(model)

        # per-module saved activations / output-grads (cleared every step)
        self.m_a: Dict[nn.Module, torch.Tensor] = {}
        self.m_g: Dict[nn.Module, torch.Tensor] = {}

        # per-module state views (into FlatBuckets, filled by _init_state)
        self.m_A: Dict[nn.Module, torch.Tensor] = {}
        self.m_G: Dict[nn.Module, torch.Tensor] = {}
        self.m_precon_grad: Dict[nn.Module, torch.Tensor] = {}

        self.module_ranks: Optional[Dict[nn.Module, Tuple[int, int]]] = None

        self.eps = 1e-10  # eigenvalue clamp (reference :115)
        self.steps = 0
        self._state_ready = False

    # ------------------------------------------------------------------ comm
    @property
    def comm(self):
        return comm_mod.get_comm()

    def _native_comm(self):
        """Opt-in (KFAC_NATIVE_COMM=1) multi-stream RCCL communicator
        for the bucket collectives: duplicate RCCL comms on their own
        HIP streams so broadcasts rooted at different owners ride
        different xGMI links (ops/csrc_rccl).  Lazily and collectively
        created at first use; None when disabled / CPU / world 1."""
        if not hasattr(self, "_native_comm_obj"):
            self._native_comm_obj = None
            import os
            if (os.environ.get("KFAC_NATIVE_COMM") == "1"
                    and torch.cuda.is_available()
                    and self.comm.size() > 1):
                from kfac_pytorch_amd.parallel.native import (
                    NativeCommunicator, native_available)
                if native_available():
                    self._native_comm_obj = NativeCommunicator.create(
                        num_comms=min(4, self.comm.size()))
        return self._native_comm_obj

    # ----------------------------------------------------------------- hooks
    def set_hook_enabled(self, mode: bool = True):
        self.hook_enabled = mode

    def _save_input_enabled(self) -> bool:
        return (self.hook_enabled and torch.is_grad_enabled()
                and self.steps % self.fac_update_freq == 0)

    def _save_grad_enabled(self) -> bool:
        return self.hook_enabled and self.steps % self.fac_update_freq == 0

    def _forward_hook_event(self, module, input):
        """Save the module input (a) by reference."""
        if self._save_input_enabled():
            self.m_a[module] = input[0].data
            self._overlap_factor(module, "A", input[0].data)

    def _backward_hook_event(self, module, grad_input, grad_output):
        """Save the grad wrt output (g) by reference."""
        if self._save_grad_enabled():
            self.m_g[module] = grad_output[0].data
            self._overlap_factor(module, "G", grad_output[0].data)

    # ------------------------------------------ factor/backward overlap
    def _overlap_enabled(self) -> bool:
        """KFAC_FACTOR_OVERLAP (default on, GPU only): compute each
        layer's FRESH factor on a side stream the moment its hook
        fires, so the batch-sized factor products (im2col + MFMA SYRK)
        hide under the rest of forward/backward instead of running as
        a serial phase after it.  ``step()`` then only applies the
        running-average AXPY.  Numerics identical to the inline path
        (fresh factor staged, last capture wins on hook refires)."""
        if not hasattr(self, "_ov_flag"):
            import os
            self._ov_flag = (os.environ.get("KFAC_FACTOR_OVERLAP", "1")
                             != "0" and torch.cuda.is_available())
        return self._ov_flag

    def _overlap_factor(self, module, kind, tensor):
        if not self._overlap_enabled() or not tensor.is_cuda:
            return
        if not hasattr(self, "_ov_stream"):
            self._ov_stream = torch.cuda.Stream()
            self._ov_ready = torch.cuda.Event()
            self._ov_fresh: Dict = {}   # (module, kind) -> staging
            self._ov_new = set()        # keys computed since last consume
        key = (module, kind)
        self._ov_ready.record()
        self._ov_stream.wait_event(self._ov_ready)
        with torch.cuda.stream(self._ov_stream):
            # the capture is allocated on the default stream; pin it
            # until the side-stream factor product finishes
            tensor.record_stream(self._ov_stream)
            stg = self._ov_fresh.get(key)
            if kind == "A":
                res = self.computeA(tensor, module, out=stg, decay=None)
            else:
                res = self.computeG(tensor, module, batch_averaged=True,
                                    out=stg, decay=None)
            self._ov_fresh[key] = res
        self._ov_new.add(key)

    def _consume_overlapped(self):
        """Keys side-computed since the last call; the default stream
        is ordered behind the side stream before returning."""
        new = getattr(self, "_ov_new", None)
        if not new:
            return frozenset()
        torch.cuda.current_stream().wait_stream(self._ov_stream)
        out = frozenset(new)
        new.clear()
        return out

    # --------------------------------------------- hipGraph phase capture
    def _run_graphed(self, key, fn, fingerprint):
        """Run ``fn`` (a launch-bound phase writing only into static
        buffers) through a captured hipGraph, replaying on subsequent
        steps.  ~300 small launches collapse into one graph launch.

        ``fingerprint`` must cover everything baked into the capture:
        scalar hyperparameters (damping) and the data_ptrs of every
        input tensor whose STORAGE could be swapped between steps
        (p.grad under zero_grad(set_to_none=True)) -- any change
        triggers a clean recapture.  KFAC_PRED_GRAPH=0 disables."""
        import os
        if (not torch.cuda.is_available()
                or os.environ.get("KFAC_PRED_GRAPH", "1") == "0"
                or getattr(self, "_graph_disabled", False)):
            fn()
            return
        if not hasattr(self, "_graphs"):
            self._graphs: Dict = {}
        entry = self._graphs.get(key)
        if entry is not None and entry[1] == fingerprint:
            entry[0].replay()
            return
        try:
            import warnings
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                fn()  # warmup (allocator state, lazy inits)
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with warnings.catch_warnings(record=True) as caught:
                warnings.simplefilter("always")
                with torch.cuda.graph(graph):
                    fn()
            # torch only WARNS when a capture lands empty; an empty
            # graph would replay as a silent no-op forever -- treat it
            # as a capture failure and stay eager
            if any("empty" in str(w.message).lower() for w in caught):
                raise RuntimeError("empty hipGraph capture")
            self._graphs[key] = (graph, fingerprint)
            graph.replay()
        except Exception:
            # capture-unfriendly op somewhere: stay eager for good
            self._graph_disabled = True
            fn()

    def _register_module_hooks(self, model: nn.Module):
        name_idx = 0
        for module in model.modules():
            classname = module.__class__.__name__
            if classname not in SUPPORTED_MODULES:
                continue
            if (self.exclude_vocabulary_size is not None
                    and classname == 'Linear'
                    and module.out_features == self.exclude_vocabulary_size):
                continue  # exclude pre-softmax vocab projection (ref :139-140)
            # grouped convs get exact block-diagonal factors (one
            # block per group; beyond the reference, which computes a
            # single WRONG dense factor for groups > 1)
            if (classname == 'Linear' and module.out_features >= 10000
                    and self.exclude_vocabulary_size is None):
                logger.warning(
                    "KFAC: hooking Linear with out_features=%d -- its G "
                    "factor alone is %d MB fp32; pass "
                    "exclude_vocabulary_size=%d to skip vocab-sized "
                    "projections (reference behavior for LM heads)",
                    module.out_features,
                    module.out_features ** 2 * 4 // 2 ** 20,
                    module.out_features)
            self.modules.append(module)
            self._hook_handles.append(
                module.register_forward_pre_hook(self._forward_hook_event))
            self._hook_handles.append(
                module.register_full_backward_hook(self._backward_hook_event))
            self.module_names.append(f'module_name_{classname}_{name_idx}')
            name_idx += 1
        if comm_mod.is_initialized() and self.comm.rank() == 0:
            logger.info("#register modules: %s", len(self.modules))

    # ------------------------------------------------------------- scheduling
    def schedule_module_ranks(self):
        raise NotImplementedError

    def _round_robin_ranks(self, factor_wise: bool = False):
        """Layer->rank assignment.

        Default is COST-AWARE (LPT greedy bin-packing by the O(m^3)
        inverse/eigensolve cost of each layer's factors): the
        reference's plain round-robin (kfac/kfac_preconditioner_inv.py
        :62-77) puts whole 4608-dim conv factors on whichever rank the
        rotation lands them, so at 8 GPUs the step is bound by the
        straggler (~2x the balanced share for ResNet-50).  Any
        assignment is numerically equivalent -- ownership is pure
        scheduling -- so balancing is free throughput at scale;
        KFAC_SCHEDULE=roundrobin restores the reference order.  The
        factor-wise variant (world > #layers) keeps the reference's
        rank_g = rank_a + 1 scheme (kfac_preconditioner_eigen.py:75-94).
        """
        import os
        module_ranks = {}
        size = self.comm.size()
        policy = os.environ.get("KFAC_SCHEDULE", "lpt")
        if factor_wise or policy == "roundrobin" or size == 1:
            rank_iter = 0
            for module in self.modules:
                rank_a = rank_iter % size
                if factor_wise:
                    rank_iter += 1
                    rank_g = rank_iter % size
                else:
                    rank_g = rank_a
                module_ranks[module] = (rank_a, rank_g)
                rank_iter += 1
        else:
            from kfac_pytorch_amd.ops.factors import (factor_dims,
                                                      factor_groups)
            costs = []
            for i, m in enumerate(self.modules):
                da, dg = factor_dims(m)
                costs.append((factor_groups(m) * (da ** 3 + dg ** 3), i))
            # LPT: heaviest first into the currently lightest bin;
            # deterministic (stable sort + index tiebreak) so every
            # rank computes the identical schedule
            costs.sort(key=lambda c: (-c[0], c[1]))
            loads = [0] * size
            for cost, i in costs:
                r = min(range(size), key=lambda k: (loads[k], k))
                loads[r] += cost
                module_ranks[self.modules[i]] = (r, r)
        self.module_ranks = module_ranks
        if self.comm.rank() == 0:
            logger.info('module_ranks: %s', list(module_ranks.values()))
        return module_ranks

    # ------------------------------------------------------- state allocation
    def _init_state(self):
        """Allocate flat-bucketed per-module state. Subclasses extend."""
        raise NotImplementedError

    def _reference_param(self) -> torch.Tensor:
        return self.param_groups[0]['params'][0]

    def _state_device(self) -> torch.device:
        return self._reference_param().device

    # ----------------------------------------------- flat-bucket allocation
    def _alloc_factor_buckets(self, owner_only: bool = False):
        """Allocate m_A/m_G as views of ONE flat fp32 buffer (identity-
        initialized, reference inits A=G=I at step 0:
        kfac/kfac_preconditioner_inv.py:84-90).

        ``owner_only=True`` (DP variants) allocates only the factors this
        rank owns -- DP-KFAC never communicates factors so nobody else
        needs storage (kfac/kfac_preconditioner_inv_dp.py:75-95).
        """
        from kfac_pytorch_amd.ops.factors import factor_dims
        device = self._state_device()
        rank = self.comm.rank()
        self.factor_bucket = FlatBucket(torch.float32, device)
        for i, m in enumerate(self.modules):
            rank_a, rank_g = self.module_ranks[m]
            da, dg = factor_dims(m)
            if not owner_only or rank == rank_a:
                self.factor_bucket.add(
                    f"A{i}", torch.Size(self._block_shape(m, da)))
            if not owner_only or rank == rank_g:
                self.factor_bucket.add(
                    f"G{i}", torch.Size(self._block_shape(m, dg)))
        self.factor_bucket.freeze(0.0)
        for i, m in enumerate(self.modules):
            if f"A{i}" in self.factor_bucket:
                v = self.factor_bucket.view(f"A{i}")
                v.diagonal(dim1=-2, dim2=-1).fill_(1.0)
                self.m_A[m] = v
            if f"G{i}" in self.factor_bucket:
                v = self.factor_bucket.view(f"G{i}")
                v.diagonal(dim1=-2, dim2=-1).fill_(1.0)
                self.m_G[m] = v

    def _alloc_owner_buckets(self, specs) -> List[FlatBucket]:
        """Build one FlatBucket per owner rank from
        ``specs = [(name, shape, owner_rank), ...]`` and return the list.
        Every rank allocates every bucket (broadcast destinations); views
        are fetched via ``_owner_view``."""
        device = self._state_device()
        size = self.comm.size()
        buckets = [FlatBucket(torch.float32, device) for _ in range(size)]
        for name, shape, owner in specs:
            buckets[owner].add(name, torch.Size(shape))
        for b in buckets:
            b.freeze(0.0)
        return buckets

    @staticmethod
    def _owner_view(buckets: List[FlatBucket], name: str) -> torch.Tensor:
        for b in buckets:
            if name in b:
                return b.view(name)
        raise KeyError(name)

    # -------------------------------------------------------- bucketed comm
    def _allreduce_bucket_avg(self, bucket: FlatBucket):
        """One allreduce-average for a whole phase's tensors."""
        if bucket.buffer is None or bucket.buffer.numel() == 0:
            return
        nat = self._native_comm()
        if nat is not None:
            nat.all_reduce(bucket.buffer, average=True)
            nat.join()
            return
        self.comm.allreduce(bucket.buffer, op=self.comm.Average)

    def _broadcast_owner_buckets(self, buckets: List[FlatBucket]):
        """Async broadcast of each owner's flat bucket from its rank,
        issued concurrently on rotating duplicate communicators so
        different roots ride different xGMI links; then drain.

        Replaces the reference's per-layer broadcast bursts
        (kfac/kfac_preconditioner_inv.py:132-142,164-175)."""
        nat = self._native_comm()
        if nat is not None:
            for r, b in enumerate(buckets):
                if b.buffer is None or len(b) == 0:
                    continue
                nat.broadcast(b.buffer, root=r)
            nat.join()  # stream-ordered, no host block
            return
        c = self.comm
        c.ensure_rotating_groups()
        handles = []
        for r, b in enumerate(buckets):
            if b.buffer is None or len(b) == 0:
                continue
            handles.append(c.broadcast_async_(b.buffer, src=r,
                                              group=c.rotating_group(r)))
        c.synchronize(handles)

    # ------------------------------------------------ phase methods (virtual)
    def _compute_factors(self):
        raise NotImplementedError

    def _communicate_factors(self):
        raise NotImplementedError

    def _compute_inverse(self):
        raise NotImplementedError

    def _communicate_inverse(self):
        raise NotImplementedError

    def _compute_pred(self):
        raise NotImplementedError

    def _communicate_pred(self):
        raise NotImplementedError

    # --------------------------------------------------------- grad plumbing
    @staticmethod
    def _block_shape(module, d: int, vec: bool = False):
        """Factor-state shape: (d,)/(d,d), or group-stacked for grouped
        convs (block-diagonal K-FAC)."""
        from kfac_pytorch_amd.ops.factors import factor_groups
        g = factor_groups(module)
        if vec:
            return (g, d) if g > 1 else (d,)
        return (g, d, d) if g > 1 else (d, d)

    def _get_grad(self, module: nn.Module) -> torch.Tensor:
        """Gradient as [out_dim, in_dim(+1)] -- group-stacked
        [g, out/g, in/g*kh*kw(+1)] for grouped convs (reference
        kfac/kfac_preconditioner_inv.py:145-154)."""
        if isinstance(module, nn.Conv2d) and module.groups > 1:
            g = module.groups
            grad = module.weight.grad.data.view(
                g, module.out_channels // g, -1)
            if module.bias is not None:
                grad = torch.cat(
                    [grad, module.bias.grad.data.view(g, -1, 1)], 2)
            return grad
        if isinstance(module, nn.Conv2d):
            grad = module.weight.grad.data.view(
                module.weight.grad.data.size(0), -1)
        else:
            grad = module.weight.grad.data
        if module.bias is not None:
            grad = torch.cat([grad, module.bias.grad.data.view(-1, 1)], 1)
        return grad

    def _reshape_preconditioned_grad(self, module, v: torch.Tensor):
        """Split [out, in(+1)] (or the group-stacked 3-D form) back into
        weight/bias shapes (reference kfac_preconditioner_inv.py:178-186)."""
        if v.dim() == 3:
            if module.bias is not None:
                vw = v[:, :, :-1].reshape(module.weight.grad.data.size())
                vb = v[:, :, -1].reshape(module.bias.grad.data.size())
                return [vw, vb]
            return [v.reshape(module.weight.grad.data.size())]
        if module.bias is not None:
            vw = v[:, :-1].reshape(module.weight.grad.data.size())
            vb = v[:, -1:].reshape(module.bias.grad.data.size())
            return [vw, vb]
        return [v.reshape(module.weight.grad.data.size())]

    def _update_grad_in_place(self):
        """Copy preconditioned grads into .grad and apply the kl-clip
        rescale nu = min(1, sqrt(kl_clip/|sum(v*g*lr^2)|))
        (reference: kfac/kfac_preconditioner_inv.py:188-217) with no
        host-device sync: vg_sum stays a device scalar."""
        use_clip = self.kl_clip is not None
        vg_sum = None
        grads: List[torch.Tensor] = []
        for module in self.modules:
            v = self._reshape_preconditioned_grad(
                module, self.m_precon_grad[module])
            params = [module.weight] + (
                [module.bias] if module.bias is not None else [])
            for p, vi in zip(params, v):
                if use_clip:
                    contrib = (vi * p.grad.data).sum() * (self.lr ** 2)
                    vg_sum = contrib if vg_sum is None else vg_sum + contrib
                p.grad.data.copy_(vi)
                grads.append(p.grad.data)

        if use_clip and vg_sum is not None:
            if self.exclude_communicate_inverse:
                return  # nu == 1 (reference :209-212)
            nu = torch.clamp(
                (self.kl_clip / vg_sum.abs().clamp_min(1e-30)).sqrt(),
                max=1.0)
            torch._foreach_mul_(grads, nu)

    # ------------------------------------------------------------------ step
    def _phase(self, name: str):
        """Optional per-phase wall timing (KFAC_PHASE_TIMING=1): syncs the
        GPU around each phase and accumulates seconds into
        ``self.phase_times`` for perf attribution."""
        import contextlib
        import os
        import time
        if not os.environ.get("KFAC_PHASE_TIMING"):
            return contextlib.nullcontext()

        @contextlib.contextmanager
        def timer():
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            yield
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            if not hasattr(self, "phase_times"):
                self.phase_times = {}
            self.phase_times[name] = self.phase_times.get(name, 0.0) + \
                time.perf_counter() - t0
        return timer()

    def step(self, closure=None, epoch=None):
        """One K-FAC step (4-phase template, reference
        kfac/kfac_preconditioner_base.py:185-230)."""
        group = self.param_groups[0]
        self.lr = group['lr']
        self.damping = group['damping']
        self.fac_update_freq = group['fac_update_freq']
        self.kfac_update_freq = group['kfac_update_freq']

        if self.module_ranks is None:
            self.schedule_module_ranks()
        if not self._state_ready:
            self._init_state()
            self._state_ready = True
            # create the rotating duplicate communicators EAGERLY and
            # collectively here (every rank passes this point at step 0
            # with identical state) instead of lazily inside the first
            # broadcast -- group creation is itself a collective, and a
            # lazy first-use site is a hang hazard if any rank's call
            # pattern ever diverged
            if (self.comm.size() > 1 and self._native_comm() is None
                    and not self.exclude_communicate_inverse):
                self.comm.ensure_rotating_groups()

        if self.steps % self.fac_update_freq == 0:
            if not self.exclude_compute_factor:
                with self._phase("compute_factor"):
                    self._compute_factors()
            if not self.exclude_communicate_factor and self.comm.size() > 1:
                with self._phase("comm_factor"):
                    self._communicate_factors()

        if self.steps % self.kfac_update_freq == 0:
            if not self.exclude_compute_inverse:
                # surface any rocSOLVER failure from the PREVIOUS update
                # before issuing new solves (deferred to keep the solve
                # path sync-free; the queue is drained here anyway)
                from kfac_pytorch_amd.ops.linalg import check_deferred_info
                check_deferred_info()
                with self._phase("compute_inverse"):
                    self._compute_inverse()
            if (not self.exclude_communicate_inverse
                    and self.communicate_inverse_or_not
                    and self.comm.size() > 1):
                with self._phase("comm_inverse"):
                    self._communicate_inverse()

        if not self.exclude_compute_inverse:
            with self._phase("compute_pred"):
                self._compute_pred()

        if (not self.exclude_communicate_inverse
                and not self.communicate_inverse_or_not
                and self.comm.size() > 1):
            with self._phase("comm_pred"):
                self._communicate_pred()

        if not self.exclude_compute_inverse:
            with self._phase("update_grad"):
                self._update_grad_in_place()

        self.steps += 1
        self.m_a, self.m_g = {}, {}


def _iter_buckets(pre):
    """All FlatBuckets a preconditioner allocated, keyed by name."""
    out = {}
    if getattr(pre, "factor_bucket", None) is not None:
        out["factor"] = [pre.factor_bucket]
    for attr in ("inv_buckets", "pred_buckets", "eig_buckets"):
        if getattr(pre, attr, None) is not None:
            out[attr] = list(getattr(pre, attr))
    return out


def kfac_state_dict(pre) -> dict:
    """Serializable K-FAC state: step counter + every flat state buffer.

    The reference never checkpoints K-FAC state (factors rebuild from
    identity after restart -- SURVEY.md S5 'Checkpoint / resume'); this
    makes preconditioning resume warm.  Valid to reload only with the
    same model / world size / schedule.
    """
    state = {"steps": pre.steps, "buckets": {}}
    for name, buckets in _iter_buckets(pre).items():
        state["buckets"][name] = [
            (b.buffer.detach().cpu() if b.buffer is not None else None)
            for b in buckets]
    return state


def load_kfac_state_dict(pre, state: dict) -> None:
    """Restore state saved by :func:`kfac_state_dict` (allocates the
    buckets first if the preconditioner has not stepped yet)."""
    if not pre._state_ready:
        if pre.module_ranks is None:
            pre.schedule_module_ranks()
        pre._init_state()
        pre._state_ready = True
    pre.steps = int(state["steps"])
    current = _iter_buckets(pre)
    for name, buffers in state["buckets"].items():
        if name not in current:
            raise KeyError(f"checkpoint has bucket group {name!r} the "
                           "preconditioner did not allocate (different "
                           "algorithm or world size?)")
        buckets = current[name]
        if len(buckets) != len(buffers):
            raise ValueError(f"bucket group {name!r}: checkpoint has "
                             f"{len(buffers)} buffers, expected "
                             f"{len(buckets)}")
        for b, saved in zip(buckets, buffers):
            if saved is None or b.buffer is None:
                continue
            if b.buffer.numel() != saved.numel():
                raise ValueError(f"bucket group {name!r}: size mismatch "
                                 "(different model/schedule?)")
            b.buffer.copy_(saved.to(b.buffer.device))


class KFACParamScheduler:
    """Epoch-schedule for damping and update frequencies
    (reference: kfac/kfac_preconditioner_base.py:233-301)."""

    def __init__(self, kfac, damping_alpha=1, damping_schedule=None,
                 update_freq_alpha=1, update_freq_schedule=None,
                 start_epoch=0):
        self.kfac = kfac
        params = self.kfac.param_groups[0]
        self.damping_base = params['damping']
        self.damping_alpha = damping_alpha
        self.damping_schedule = damping_schedule
        self.damping_factor_func = self._get_factor_func(
            damping_schedule, damping_alpha)
        self.fac_update_freq_base = params['fac_update_freq']
        self.kfac_update_freq_base = params['kfac_update_freq']
        self.update_freq_alpha = update_freq_alpha
        self.update_freq_schedule = update_freq_schedule
        self.update_freq_factor_func = self._get_factor_func(
            update_freq_schedule, update_freq_alpha)
        self.epoch = start_epoch

    @staticmethod
    def _get_factor_func(schedule, alpha):
        if schedule is not None:
            schedule = sorted(schedule, reverse=True)
        else:
            schedule = []

        def factor_func(epoch):
            factor = 1.0
            for e in schedule:
                if epoch >= e:
                    factor *= alpha
            return factor

        return factor_func

    def step(self, epoch=None):
        if epoch is not None:
            self.epoch = epoch
        else:
            self.epoch += 1
        params = self.kfac.param_groups[0]
        params['damping'] = self.damping_base * \
            self.damping_factor_func(self.epoch)
        factor = self.update_freq_factor_func(self.epoch)
        # clamp to >= 1: with update_freq_alpha < 1, int(base * factor)
        # can reach 0 and crash the next ``steps % freq`` (the reference
        # has this bug, kfac_preconditioner_base.py:288-301 -- fixed,
        # not copied)
        params['fac_update_freq'] = max(
            1, int(self.fac_update_freq_base * factor))
        params['kfac_update_freq'] = max(
            1, int(self.kfac_update_freq_base * factor))
