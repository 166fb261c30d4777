from kfac_pytorch_amd.parallel import comm
from kfac_pytorch_amd.parallel.comm import (FlatBucket, Ops, TorchCommBackend,
                                            get_comm, init, is_initialized)

__all__ = ["comm", "FlatBucket", "Ops", "TorchCommBackend", "get_comm",
           "init", "is_initialized"]
