"""Loader for the in-tree CDNA4 HIP extension (`_kfac_hip`).

The extension is built in-tree (``python setup.py build_ext --inplace``
or ``__graft_entry__.build()``) for gfx950 only.  On a GPU box the HIP
path is mandatory: any op called with CUDA tensors raises loudly if the
extension is missing rather than silently falling back to eager torch.
CPU tensors never touch this module (the torch implementations in
``factors.py``/``linalg.py`` are the CPU path and the numerics oracle).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

_mod = None
_load_error: Optional[str] = None


def _load():
    global _mod, _load_error
    if _mod is not None:
        return _mod
    try:
        from kfac_pytorch_amd.ops import _kfac_hip  # built .so, in-tree
        _mod = _kfac_hip
    except ImportError as e:  # pragma: no cover - GPU-box path
        _load_error = str(e)
        raise RuntimeError(
            "kfac_pytorch_amd HIP extension (_kfac_hip) is not built but a "
            "GPU op was requested. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
            f" Original import error: {e}"
        ) from e
    return _mod


def available() -> bool:
    try:
        _load()
        return True
    except RuntimeError:
        return False


_solver_mod = None


def load_solver():
    """Import the in-tree async/batched rocSOLVER extension
    (``_kfac_solver``); raises RuntimeError if it is not built."""
    global _solver_mod
    if _solver_mod is None:
        try:
            from kfac_pytorch_amd.ops import _kfac_solver  # built .so
            _solver_mod = _kfac_solver
        except ImportError as e:  # pragma: no cover - GPU-box path
            raise RuntimeError(
                "kfac_pytorch_amd solver extension (_kfac_solver) is not "
                "built. Build in-tree with `python setup.py build_ext "
                f"--inplace`. Original import error: {e}"
            ) from e
    return _solver_mod


def has_solver() -> bool:
    try:
        return load_solver() is not None
    except RuntimeError:
        return False


_rccl_mod = None


def load_rccl():
    """Import the in-tree multi-stream RCCL communicator extension
    (``_kfac_rccl``); raises RuntimeError if it is not built."""
    global _rccl_mod
    if _rccl_mod is None:
        try:
            from kfac_pytorch_amd.ops import _kfac_rccl  # built .so, in-tree
            _rccl_mod = _kfac_rccl
        except ImportError as e:  # pragma: no cover - GPU-box path
            raise RuntimeError(
                "kfac_pytorch_amd RCCL extension (_kfac_rccl) is not built. "
                "Build in-tree with `python setup.py build_ext --inplace` "
                f"(PYTORCH_ROCM_ARCH=gfx950). Original import error: {e}"
            ) from e
    return _rccl_mod


# -- fused elementwise: V /= (dG dA^T + damping) ----------------------------
def eigen_scale_(v: torch.Tensor, dG: torch.Tensor, dA: torch.Tensor,
                 damping: float) -> torch.Tensor:
    return _load().eigen_scale_(v, dG, dA, float(damping))


def eigen_scale_batched_(v: torch.Tensor, dG: torch.Tensor,
                         dA: torch.Tensor, damping: float) -> None:
    """In-place batched V[b] /= (dG[b] dA[b]^T + damping)."""
    _load().eigen_scale_batched_(v, dG, dA, float(damping))


# -- MFMA SYRK factor kernel -------------------------------------------------
def syrk_factor_(x: torch.Tensor, out: torch.Tensor, row_scale: float,
                 denom: float, bias: bool, decay: float) -> torch.Tensor:
    """out = (1-decay)*out + decay * (s*[x|1])^T (s*[x|1]) / denom.

    decay < 0 means plain overwrite (no running average).
    """
    return _load().syrk_factor_(x, out, float(row_scale), float(denom),
                                bool(bias), float(decay))


def has_syrk() -> bool:
    try:
        return hasattr(_load(), "syrk_factor_")
    except RuntimeError:
        return False


# -- im2col patch extraction -------------------------------------------------
def im2col(x: torch.Tensor, kh: int, kw: int, sh: int, sw: int,
           ph: int, pw: int, dh: int = 1, dw: int = 1) -> torch.Tensor:
    return _load().im2col(x, kh, kw, sh, sw, ph, pw, dh, dw)


def has_im2col() -> bool:
    try:
        return hasattr(_load(), "im2col")
    except RuntimeError:
        return False


# -- batched Jacobi symmetric eigensolver ------------------------------------
def jacobi_eigh_batched(mats):
    """Batched symmetric eigensolve of fp32 GPU matrices (one workgroup
    each, all in one launch). Returns a list of (w, V) pairs; eigenvalues
    UNSORTED (Jacobi diagonal order)."""
    mod = _load()
    W, V = mod.jacobi_eigh_batched(list(mats))
    out = []
    woff = moff = 0
    for a in mats:
        m = a.size(0)
        out.append((W[woff:woff + m],
                    V[moff:moff + m * m].view(m, m)))
        woff += m
        moff += m * m
    return out


def jacobi_eigh(x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Single-matrix Jacobi eigensolve, eigenvalues sorted ascending
    (torch.linalg.eigh contract)."""
    (w, V), = jacobi_eigh_batched([x])
    w, idx = torch.sort(w)
    return w, V[:, idx].contiguous()


def has_jacobi_eigh(m: int) -> bool:
    try:
        mod = _load()
    except RuntimeError:
        return False
    return (hasattr(mod, "jacobi_eigh_batched")
            and m <= int(mod.jacobi_eigh_max_dim()))
