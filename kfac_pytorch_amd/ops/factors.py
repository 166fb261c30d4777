"""Kronecker-factor math for K-FAC: A (activation) and G (grad-output) covariances.

Numerics follow the reference formulas exactly
(reference: kfac/utils.py:73-140):

* ``Linear`` A: 3-D activations are **averaged over the sequence dim**
  (kfac/utils.py:98-99), a bias column of ones is appended, and
  ``A = a^T (a / B)``.
* ``Conv2d`` A: im2col patches ``(B, oh, ow, in_c*kh*kw)``
  (kfac/utils.py:33-54), bias column appended, everything divided by the
  spatial size, then ``A = a^T (a / B)`` (kfac/utils.py:86-94).
* ``Conv2d`` G: grad-outputs scaled by ``B * spatial`` (batch-averaged
  loss), flattened to rows, ``G = g^T (g / rows)`` (kfac/utils.py:118-129).
* ``Linear`` G: seq-mean, then ``G = g^T (g * B)`` when batch-averaged
  (kfac/utils.py:131-140).

The single compute primitive behind all four cases is
:func:`sym_factor`, ``F = (s*x)^T (s*x) / denom`` with an optional fused
bias column -- a symmetric rank-k (SYRK) product.  On MI355X GPUs it
dispatches to a hand-written MFMA HIP kernel (bf16 inputs, fp32
accumulate, fused bias column / scaling / running-average epilogue); the
pure-torch path below is the CPU/oracle reference for it.

Unlike the reference, the bias-augmented matrix is never materialized
(the reference ``torch.cat``s a ones column onto the full im2col matrix,
kfac/utils.py:92,102 -- a full extra copy of the largest tensor in the
step); the augmented factor is assembled blockwise instead.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = [
    "extract_patches",
    "sym_factor",
    "update_running_avg",
    "ComputeA",
    "ComputeG",
    "factor_dims",
]


def extract_patches(x: torch.Tensor, kernel_size, stride, padding,
                    dilation=(1, 1)) -> torch.Tensor:
    """im2col: (B, C, H, W) -> (B*oh*ow, C*kh*kw) patch rows.

    Row-major patch ordering matches the flattened Conv2d weight layout
    ``[out, in*kh*kw]`` (in, kh, kw fastest-last), i.e. the same ordering
    the reference's double-unfold produces (kfac/utils.py:33-54).

    GPU path: the HIP im2col kernel emits **bf16** patch rows directly
    (the MFMA SYRK kernel's input dtype -- capture is bf16, accumulation
    fp32, per the framework's GPU numerics policy). CPU path stays fp32.
    """
    if x.is_cuda:
        from kfac_pytorch_amd.ops import _ext
        return _ext.im2col(x.contiguous(), kernel_size[0], kernel_size[1],
                           stride[0], stride[1], padding[0], padding[1],
                           dilation[0], dilation[1])
    cols = F.unfold(x, kernel_size=kernel_size, stride=stride,
                    padding=padding, dilation=dilation)  # (B, C*kh*kw, L)
    return cols.transpose(1, 2).reshape(-1, cols.size(1))


def sym_factor(x: torch.Tensor, *, row_scale: float = 1.0, denom: float = 1.0,
               bias: bool = False, out: Optional[torch.Tensor] = None,
               decay: Optional[float] = None) -> torch.Tensor:
    """F = (row_scale * [x | 1])^T (row_scale * [x | 1]) / denom  (fp32).

    With ``bias=True`` the implicit ones-column is scaled by ``row_scale``
    too, matching the reference order (ones appended *before* the spatial
    division, kfac/utils.py:92-93).

    With ``out``+``decay`` the running average
    ``out = (1-decay)*out + decay*F`` is fused (reference
    update_running_avg, kfac/utils.py:66-71).

    This function is the contract the MFMA HIP kernel implements; the
    torch path here is its CPU / numerics oracle.
    """
    if x.dim() != 2:
        raise ValueError(f"sym_factor expects 2-D input, got {tuple(x.shape)}")

    if x.is_cuda:
        # MI355X path: hand-written MFMA SYRK kernel, bf16 inputs / fp32
        # accumulate, fused bias column + scale + running average.
        # Raises loudly if the gfx950 extension is not built.
        from kfac_pytorch_amd.ops import _ext
        xb = x if x.dtype == torch.bfloat16 else x.bfloat16()
        n = x.shape[1] + (1 if bias else 0)
        if out is None:
            out = torch.empty(n, n, device=x.device, dtype=torch.float32)
            dec = -1.0
        else:
            dec = -1.0 if decay is None else float(decay)
        return _ext.syrk_factor_(xb.contiguous(), out, float(row_scale),
                                 float(denom), bool(bias), dec)

    x32 = x.float()
    rows, d = x32.shape
    n = d + 1 if bias else d
    scale = (row_scale * row_scale) / denom

    F_ = x32.new_empty((n, n))
    torch.mm(x32.t(), x32, out=F_[:d, :d])
    if bias:
        cs = x32.sum(dim=0)
        F_[:d, d] = cs
        F_[d, :d] = cs
        F_[d, d] = float(rows)
    F_.mul_(scale)

    if out is not None:
        if decay is None:
            out.copy_(F_)
        else:
            out.mul_(1.0 - decay).add_(F_, alpha=decay)
        return out
    return F_


def update_running_avg(new: torch.Tensor, current: torch.Tensor,
                       alpha: float) -> None:
    """current = alpha*new + (1-alpha)*current, in place
    (reference: kfac/utils.py:66-71)."""
    current.mul_(1.0 - alpha)
    current.add_(new, alpha=alpha)


def _seq_mean(t: torch.Tensor) -> torch.Tensor:
    """Average every dim between batch and feature (kfac/utils.py:98-99).

    The reference handles 3-D ``(B, seq, d)``; we also fold deeper shapes
    ``(B, *, d)`` the same way so factor dims stay ``d x d`` regardless of
    sequence length (the reference's entire long-context strategy --
    SURVEY.md S5 'Long-context').
    """
    if t.dim() <= 2:
        return t
    return t.mean(dim=tuple(range(1, t.dim() - 1)))


def sym_factor_grouped(x: torch.Tensor, groups: int, *,
                       row_scale: float = 1.0, denom: float = 1.0,
                       bias: bool = False,
                       out: Optional[torch.Tensor] = None,
                       decay: Optional[float] = None) -> torch.Tensor:
    """Stacked per-group factors (g, d, d) from channel-group-major
    rows ``x (rows, g*d0)`` -- exact block-diagonal K-FAC for grouped
    convolutions (one batched bmm; each block identical in math to
    :func:`sym_factor` on that group's column slice).

    The reference computes a single WRONG dense factor for groups > 1
    (its weight reshape no longer matches the patch layout); this
    framework preconditions each group's block exactly instead.
    """
    rows, total = x.shape
    d0 = total // groups
    xg = x.float().view(rows, groups, d0).transpose(0, 1)  # (g, rows, d0)
    if bias:
        xg = torch.cat([xg, xg.new_ones(groups, rows, 1)], dim=2)
    F_ = torch.bmm(xg.transpose(1, 2), xg)
    F_.mul_((row_scale * row_scale) / denom)
    if out is not None:
        if decay is None:
            out.copy_(F_)
        else:
            out.mul_(1.0 - decay).add_(F_, alpha=decay)
        return out
    return F_


class ComputeA:
    """Kronecker factor A from a module's saved input activation."""

    def __call__(self, a: torch.Tensor, layer: nn.Module,
                 out: Optional[torch.Tensor] = None,
                 decay: Optional[float] = None) -> torch.Tensor:
        if isinstance(layer, nn.Linear):
            return self.linear(a, layer, out=out, decay=decay)
        if isinstance(layer, nn.Conv2d):
            return self.conv2d(a, layer, out=out, decay=decay)
        raise NotImplementedError(
            f"KFAC does not support layer: {layer.__class__.__name__}")

    @staticmethod
    def conv2d(a, layer, out=None, decay=None):
        B = a.size(0)
        patches = extract_patches(a, layer.kernel_size, layer.stride,
                                  layer.padding, layer.dilation)
        spatial = patches.size(0) // B
        if layer.groups > 1:
            # patch columns are channel-major, so each group's slice is
            # contiguous: one batched bmm builds every block factor
            return sym_factor_grouped(
                patches, layer.groups, row_scale=1.0 / spatial,
                denom=float(B), bias=layer.bias is not None, out=out,
                decay=decay)
        return sym_factor(patches, row_scale=1.0 / spatial, denom=float(B),
                          bias=layer.bias is not None, out=out, decay=decay)

    @staticmethod
    def linear(a, layer, out=None, decay=None):
        a = _seq_mean(a)
        B = a.size(0)
        return sym_factor(a, row_scale=1.0, denom=float(B),
                          bias=layer.bias is not None, out=out, decay=decay)


class ComputeG:
    """Kronecker factor G from a module's saved output gradient."""

    def __call__(self, g: torch.Tensor, layer: nn.Module,
                 batch_averaged: bool = True,
                 out: Optional[torch.Tensor] = None,
                 decay: Optional[float] = None) -> torch.Tensor:
        if isinstance(layer, nn.Conv2d):
            return self.conv2d(g, layer, batch_averaged, out=out, decay=decay)
        if isinstance(layer, nn.Linear):
            return self.linear(g, layer, batch_averaged, out=out, decay=decay)
        raise NotImplementedError(
            f"KFAC does not support layer: {layer.__class__.__name__}")

    @staticmethod
    def conv2d(g, layer, batch_averaged=True, out=None, decay=None):
        B = g.size(0)
        spatial = g.size(2) * g.size(3)
        rows = g.permute(0, 2, 3, 1).reshape(-1, g.size(1))
        scale = float(spatial) * (float(B) if batch_averaged else 1.0)
        if layer.groups > 1:
            return sym_factor_grouped(
                rows, layer.groups, row_scale=scale,
                denom=float(rows.size(0)), bias=False, out=out,
                decay=decay)
        return sym_factor(rows, row_scale=scale, denom=float(rows.size(0)),
                          bias=False, out=out, decay=decay)

    @staticmethod
    def linear(g, layer, batch_averaged=True, out=None, decay=None):
        g = _seq_mean(g)
        B = float(g.size(0))
        # batch_averaged: G = g^T (g*B) = B * g^T g  (row_scale=B, denom=B)
        # else:           G = g^T (g/B)              (row_scale=1, denom=B)
        row_scale = B if batch_averaged else 1.0
        return sym_factor(g, row_scale=row_scale, denom=B, bias=False,
                          out=out, decay=decay)


def factor_dims(layer: nn.Module) -> Tuple[int, int]:
    """(dim_A, dim_G) PER FACTOR BLOCK for a supported layer, including
    the bias column.  Grouped convolutions have ``factor_groups(layer)``
    independent blocks of these dims (block-diagonal K-FAC)."""
    if isinstance(layer, nn.Linear):
        da, dg = layer.in_features, layer.out_features
    elif isinstance(layer, nn.Conv2d):
        da = (layer.in_channels // layer.groups) * layer.kernel_size[0] * \
            layer.kernel_size[1]
        dg = layer.out_channels // layer.groups
    else:
        raise NotImplementedError(
            f"KFAC does not support layer: {layer.__class__.__name__}")
    if layer.bias is not None:
        da += 1
    return da, dg


def factor_groups(layer: nn.Module) -> int:
    """Number of independent factor blocks (1 except grouped convs)."""
    if isinstance(layer, nn.Conv2d):
        return int(layer.groups)
    return 1
