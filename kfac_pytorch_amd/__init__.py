"""kfac_pytorch_amd -- MI355X-native distributed K-FAC for PyTorch-ROCm.

Public API kept drop-in compatible with the reference
(reference: kfac/__init__.py:8-16, kfac/dp_kfac.py:4-39):

    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.backend as backend

    dist.init_process_group('nccl')     # RCCL over xGMI on ROCm
    backend.init("Torch")
    KFAC = kfac.get_kfac_module(kfac='eigen_dp')
    preconditioner = KFAC(model, lr=..., damping=...)
    ...
    loss.backward()          # DDP-averaged gradients
    preconditioner.step()
    optimizer.step()
"""

from kfac_pytorch_amd.preconditioner import (KFACEigen, KFACEigenDP,
                                             KFACInverse, KFACInverseDP,
                                             KFACParamScheduler)

__version__ = "0.1.0"

# reference name -> class map (kfac/__init__.py:8-13)
kfac_mappers = {
    'inverse': KFACInverse,
    'eigen': KFACEigen,
    'inverse_dp': KFACInverseDP,
    'eigen_dp': KFACEigenDP,
}

# aliases matching the reference's exported class names
KFAC_INV = KFACInverse
KFAC_EIGEN = KFACEigen
KFAC_INV_DP = KFACInverseDP
KFAC_EIGEN_DP = KFACEigenDP


def get_kfac_module(kfac: str = 'eigen_dp'):
    """Name -> preconditioner class (reference: kfac/__init__.py:15-16)."""
    return kfac_mappers[kfac]


def DP_KFAC(model, inv_type: str = 'eigen', lr=0.1, damping=0.001,
            fac_update_freq=1, kfac_update_freq=1, kl_clip=0.001,
            factor_decay=0.95, exclude_vocabulary_size=None,
            hook_enabled=True, exclude_parts=''):
    """DP-KFAC factory choosing the eigen vs inverse DP variant
    (reference: kfac/dp_kfac.py:4-39)."""
    cls = KFACEigenDP if inv_type == 'eigen' else KFACInverseDP
    return cls(model=model, lr=lr, damping=damping,
               fac_update_freq=fac_update_freq,
               kfac_update_freq=kfac_update_freq, kl_clip=kl_clip,
               factor_decay=factor_decay,
               exclude_vocabulary_size=exclude_vocabulary_size,
               hook_enabled=hook_enabled, exclude_parts=exclude_parts)


__all__ = [
    "kfac_mappers", "get_kfac_module", "DP_KFAC", "KFACParamScheduler",
    "KFACInverse", "KFACEigen", "KFACInverseDP", "KFACEigenDP",
    "KFAC_INV", "KFAC_EIGEN", "KFAC_INV_DP", "KFAC_EIGEN_DP",
]
