from kfac_pytorch_amd.ops.factors import (ComputeA, ComputeG, extract_patches,
                                          factor_dims, sym_factor,
                                          update_running_avg)
from kfac_pytorch_amd.ops.linalg import (add_diagonal_, eigen_precondition,
                                         inverse_precondition, mat_eig,
                                         mat_inv)

__all__ = [
    "ComputeA", "ComputeG", "extract_patches", "factor_dims", "sym_factor",
    "update_running_avg", "add_diagonal_", "eigen_precondition",
    "inverse_precondition", "mat_eig", "mat_inv",
]
