"""Python wrapper for the native multi-stream RCCL communicator.

The native side (``kfac_pytorch_amd.ops._kfac_rccl``) is the MI355X
equivalent of the reference's raw-NCCL ``tcmm.Communicator``
(reference: packages/tcmm/src/communicator.cpp:5-117,
packages/tcmm/src/tcmm.cpp:34-43).  This wrapper replaces the
reference's MPI bootstrap (communicator.cpp:14-15) with an exchange of
RCCL unique ids over the torch.distributed store, so no MPI dependency
exists anywhere in the stack.

Usage (after ``dist.init_process_group('nccl')``)::

    comm = NativeCommunicator.create(num_comms=4)
    comm.all_reduce(t, average=True)   # rotating (comm, stream) pair
    comm.multi_bcast(factors, outputs, eig_callback)
    comm.join()                        # torch stream waits (no host block)
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

__all__ = ["NativeCommunicator", "native_available"]


def _module():
    from kfac_pytorch_amd.ops import _ext
    return _ext.load_rccl()


def native_available() -> bool:
    """True when the ``_kfac_rccl`` extension is importable AND a GPU is
    present (RCCL communicators need a device)."""
    if not torch.cuda.is_available():
        return False
    try:
        return _module() is not None
    except Exception:
        return False


class NativeCommunicator:
    """Multi-communicator/multi-stream RCCL collectives for per-layer
    K-FAC bursts.  Thin facade over the pybind ``Communicator``."""

    def __init__(self, impl):
        self._impl = impl

    # -- bootstrap -----------------------------------------------------------
    @classmethod
    def create(cls, num_comms: int = 4,
               group: Optional[object] = None) -> "NativeCommunicator":
        """Collectively create the communicator pool.

        Rank 0 generates ``num_comms`` RCCL unique ids and every rank
        receives them through the torch.distributed store
        (``broadcast_object_list``), then each rank joins all
        ``ncclCommInitRank`` calls.  Must be called on every rank of the
        (default) group with the same ``num_comms``.
        """
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed must be initialized before "
                               "NativeCommunicator.create()")
        mod = _module()
        if mod is None:
            raise RuntimeError("_kfac_rccl extension is not available")
        rank = dist.get_rank(group)
        size = dist.get_world_size(group)
        if rank == 0:
            ids: List[bytes] = [mod.get_unique_id() for _ in range(num_comms)]
        else:
            ids = [b""] * num_comms
        payload = [ids]
        dist.broadcast_object_list(payload, src=0, group=group)
        ids = payload[0]
        impl = mod.Communicator(rank, size, ids)
        return cls(impl)

    # -- collectives (rotating comm/stream per call) -------------------------
    @property
    def rank(self) -> int:
        return self._impl.rank

    @property
    def size(self) -> int:
        return self._impl.size

    @property
    def num_comms(self) -> int:
        return self._impl.num_comms

    def all_reduce(self, tensor: torch.Tensor, average: bool = False) -> None:
        self._impl.all_reduce(tensor, average)

    def reduce(self, tensor: torch.Tensor, root: int,
               average: bool = False) -> None:
        self._impl.reduce(tensor, root, average)

    def broadcast(self, tensor: torch.Tensor, root: int) -> None:
        self._impl.broadcast(tensor, root)

    def multi_bcast(self, tensors, outputs, op,
                    min_numel: int = 512 * 512) -> None:
        """Fused compute-then-broadcast: big tensors are assigned
        round-robin to owner ranks which run ``op(input, output)`` then
        broadcast the output; small ones are computed redundantly on every
        rank (reference schedule: communicator.cpp:75-117)."""
        self._impl.multi_bcast(list(tensors), list(outputs), op, min_numel)

    # -- ordering ------------------------------------------------------------
    def synchronize(self) -> None:
        """Host-blocking drain of every comm stream."""
        self._impl.synchronize()

    def join(self) -> None:
        """Stream-ordered drain: the torch current stream waits on every
        comm stream via HIP events; the host does not block."""
        self._impl.join()


def fused_eigen_multibcast(nat: "NativeCommunicator", factors,
                           packed_out) -> None:
    """The reference's tcmm research path, natively: one fused
    compute-then-broadcast pass over a list of symmetric factors
    (reference: packages/tcmm/src/communicator.cpp:75-117 driven by
    scripts/bench_ops.py:111-146).  Each factor >= 512^2 elements is
    eigendecomposed by its round-robin owner (the per-factor callback)
    and the packed ``(n, n+1)`` output ``[Q | d]`` is broadcast from
    that owner on a rotating RCCL comm; smaller factors are computed
    redundantly on every rank (a ~64^2 broadcast on xGMI costs more
    than the redundant solve).  Call ``nat.join()`` is included."""
    from kfac_pytorch_amd.ops.linalg import mat_eig

    def op(inp, out):
        d, Q = mat_eig(inp, method="auto")
        n = inp.shape[0]
        out[:, :n].copy_(Q)
        out[:, n] = d

    nat.multi_bcast(list(factors), list(packed_out), op)
    nat.join()
