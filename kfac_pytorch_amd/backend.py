"""Reference-compatible backend shim (reference: kfac/backend.py).

Usage (same module-global-``comm`` convention as the reference; import
the module, not names from it -- ``comm`` is mutated by :func:`init`):

    import kfac_pytorch_amd.backend as backend
    dist.init_process_group(...)
    backend.init("Torch")
    backend.comm.allreduce(t)

Unlike the reference there is no Horovod backend (this framework is
torch.distributed/RCCL-native) and failures raise instead of returning
exception objects (reference quirk: kfac/backend.py:40-48).
"""

import kfac_pytorch_amd.parallel.comm as _comm_mod
from kfac_pytorch_amd.parallel.comm import Ops  # noqa: F401


def init(backend: str = "Torch", **kwargs):
    _comm_mod.init(backend, **kwargs)


def __getattr__(name):
    if name == "comm":
        return _comm_mod.comm
    raise AttributeError(name)
