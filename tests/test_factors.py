"""Factor math vs independently-written oracles.

The oracles below re-implement the reference formulas directly
(materialized bias column, explicit running average) so the framework's
block-assembled / fused implementations are checked against a different
code path.
"""

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from kfac_pytorch_amd.ops.factors import (ComputeA, ComputeG, extract_patches,
                                          factor_dims, sym_factor,
                                          update_running_avg)


def oracle_patches(x, kernel_size, stride, padding):
    """Reference-style double-unfold im2col (kfac/utils.py:33-54)."""
    if padding[0] + padding[1] > 0:
        x = F.pad(x, (padding[1], padding[1], padding[0], padding[0])).data
    x = x.unfold(2, kernel_size[0], stride[0])
    x = x.unfold(3, kernel_size[1], stride[1])
    x = x.transpose_(1, 2).transpose_(2, 3).contiguous()
    return x.view(x.size(0), x.size(1), x.size(2),
                  x.size(3) * x.size(4) * x.size(5))


def oracle_A_conv(a, layer):
    B = a.size(0)
    a = oracle_patches(a, layer.kernel_size, layer.stride, layer.padding)
    spatial = a.size(1) * a.size(2)
    a = a.view(-1, a.size(-1))
    if layer.bias is not None:
        a = torch.cat([a, a.new(a.size(0), 1).fill_(1)], 1)
    a = a / spatial
    return a.t() @ (a / B)


def oracle_A_linear(a, layer):
    if a.dim() > 2:
        a = torch.mean(a, 1)
    B = a.size(0)
    if layer.bias is not None:
        a = torch.cat([a, a.new(a.size(0), 1).fill_(1)], 1)
    return a.t() @ (a / B)


def oracle_G_conv(g, layer, batch_averaged=True):
    spatial = g.size(2) * g.size(3)
    B = g.shape[0]
    g = g.transpose(1, 2).transpose(2, 3)
    g = g.reshape(-1, g.size(-1))  # last dim is channels after transposes
    if batch_averaged:
        g = g * B
    g = g * spatial
    return g.t() @ (g / g.size(0))


def oracle_G_linear(g, layer, batch_averaged=True):
    if g.dim() > 2:
        g = torch.mean(g, 1)
    B = g.size(0)
    if batch_averaged:
        return g.t() @ (g * B)
    return g.t() @ (g / B)


@pytest.mark.parametrize("kernel,stride,padding", [
    ((3, 3), (1, 1), (1, 1)),
    ((3, 3), (2, 2), (1, 1)),
    ((1, 1), (1, 1), (0, 0)),
    ((5, 5), (1, 1), (2, 2)),
    ((7, 7), (2, 2), (3, 3)),
])
def test_extract_patches_matches_double_unfold(seeded, kernel, stride, padding):
    x = torch.randn(3, 4, 14, 14)
    ours = extract_patches(x, kernel, stride, padding)
    ref = oracle_patches(x.clone(), kernel, stride, padding)
    ref = ref.view(-1, ref.size(-1))
    assert ours.shape == ref.shape
    torch.testing.assert_close(ours, ref)


@pytest.mark.parametrize("bias", [True, False])
def test_conv_factor_A(seeded, bias):
    layer = nn.Conv2d(5, 8, 3, stride=1, padding=1, bias=bias)
    a = torch.randn(4, 5, 10, 10)
    ours = ComputeA()(a, layer)
    ref = oracle_A_conv(a, layer)
    torch.testing.assert_close(ours, ref, rtol=1e-5, atol=1e-6)
    assert ours.shape[0] == factor_dims(layer)[0]


@pytest.mark.parametrize("bias", [True, False])
@pytest.mark.parametrize("seq", [None, 7])
def test_linear_factor_A(seeded, bias, seq):
    layer = nn.Linear(6, 4, bias=bias)
    shape = (8, 6) if seq is None else (8, seq, 6)
    a = torch.randn(*shape)
    ours = ComputeA()(a, layer)
    ref = oracle_A_linear(a, layer)
    torch.testing.assert_close(ours, ref, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("batch_averaged", [True, False])
def test_conv_factor_G(seeded, batch_averaged):
    layer = nn.Conv2d(5, 8, 3, padding=1)
    g = torch.randn(4, 8, 10, 10)
    ours = ComputeG()(g, layer, batch_averaged)
    ref = oracle_G_conv(g, layer, batch_averaged)
    # scale is applied after the GEMM (one pass) instead of prescaling the
    # row matrix like the oracle -> fp reordering at the 1e-5 rel level
    torch.testing.assert_close(ours, ref, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("batch_averaged", [True, False])
@pytest.mark.parametrize("seq", [None, 5])
def test_linear_factor_G(seeded, batch_averaged, seq):
    layer = nn.Linear(6, 4)
    shape = (8, 4) if seq is None else (8, seq, 4)
    g = torch.randn(*shape)
    ours = ComputeG()(g, layer, batch_averaged)
    ref = oracle_G_linear(g, layer, batch_averaged)
    torch.testing.assert_close(ours, ref, rtol=1e-5, atol=1e-6)


def test_sym_factor_fused_running_avg(seeded):
    x = torch.randn(32, 10)
    out = torch.eye(11)
    expected = out.clone()
    fresh = sym_factor(x, row_scale=0.5, denom=32.0, bias=True)
    update_running_avg(fresh, expected, 0.95)
    sym_factor(x, row_scale=0.5, denom=32.0, bias=True, out=out, decay=0.95)
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


def test_sym_factor_bf16_input_accumulates_fp32(seeded):
    x = torch.randn(64, 16)
    f32 = sym_factor(x, denom=64.0, bias=True)
    fbf = sym_factor(x.bfloat16(), denom=64.0, bias=True)
    assert fbf.dtype == torch.float32
    # bf16 capture costs ~1e-2 relative error; factors stay fp32
    torch.testing.assert_close(fbf, f32, rtol=3e-2, atol=3e-2)


def test_factor_is_symmetric_psd(seeded):
    x = torch.randn(100, 12)
    f = sym_factor(x, denom=100.0, bias=True)
    torch.testing.assert_close(f, f.t())
    evals = torch.linalg.eigvalsh(f)
    assert evals.min() > -1e-5


def test_conv_factor_matches_naive_conv_gradient(seeded):
    """im2col rows @ flattened weight == conv output (validates that the
    patch ordering matches the Conv2d weight layout, the property the
    factor/grad reshape contract depends on; oracle pattern from
    packages/tcmm/tests/custom_conv.py)."""
    layer = nn.Conv2d(3, 6, 3, stride=2, padding=1, bias=False)
    x = torch.randn(2, 3, 9, 9)
    y = layer(x)
    patches = extract_patches(x, layer.kernel_size, layer.stride,
                              layer.padding)
    w = layer.weight.view(layer.out_channels, -1)
    y2 = (patches @ w.t()).view(2, y.size(2), y.size(3), 6).permute(0, 3, 1, 2)
    torch.testing.assert_close(y, y2, rtol=1e-4, atol=1e-5)


def test_extract_patches_asymmetric_kernel():
    """(1,7)/(7,1) factorized convs (inception family) produce correct
    patch rows on the CPU oracle path."""
    torch.manual_seed(8)
    x = torch.randn(2, 3, 9, 11)
    for ks, pad in [((1, 7), (0, 3)), ((7, 1), (3, 0)), ((3, 5), (1, 2))]:
        got = extract_patches(x, ks, (1, 1), pad)
        cols = F.unfold(x, kernel_size=ks, stride=(1, 1), padding=pad)
        ref = cols.transpose(1, 2).reshape(-1, cols.size(1))
        torch.testing.assert_close(got, ref)


def test_conv_factor_asymmetric_kernel_shapes(single_process_comm):
    """K-FAC steps through a model with (1,7)/(7,1) convs -- factor
    dims include kh*kw correctly."""
    import kfac_pytorch_amd as kfac
    m = torch.nn.Sequential(
        torch.nn.Conv2d(3, 4, (1, 7), padding=(0, 3)),
        torch.nn.ReLU(),
        torch.nn.Conv2d(4, 5, (7, 1), padding=(3, 0)),
        torch.nn.AdaptiveAvgPool2d(1), torch.nn.Flatten(),
        torch.nn.Linear(5, 3))
    pre = kfac.KFAC_EIGEN_DP(m, damping=0.01)
    from kfac_pytorch_amd.ops.factors import factor_dims
    da0, dg0 = factor_dims(m[0])
    assert (da0, dg0) == (3 * 7 + 1, 4)
    out = m(torch.randn(2, 3, 9, 9))
    F.cross_entropy(out, torch.tensor([0, 1])).backward()
    pre.step()
    assert all(torch.isfinite(p.grad).all() for p in m.parameters())
