from kfac_pytorch_amd.preconditioner.base import KFACBase, KFACParamScheduler
from kfac_pytorch_amd.preconditioner.inverse import KFACInverse
from kfac_pytorch_amd.preconditioner.eigen import KFACEigen
from kfac_pytorch_amd.preconditioner.inverse_dp import KFACInverseDP
from kfac_pytorch_amd.preconditioner.eigen_dp import KFACEigenDP

__all__ = [
    "KFACBase", "KFACParamScheduler", "KFACInverse", "KFACEigen",
    "KFACInverseDP", "KFACEigenDP",
]
