"""Property-based tests (hypothesis) for the factor math: the K-FAC
statistics formulas are convergence-critical and must hold for every
shape/stride/bias combination, not just the hand-picked cases
(SURVEY.md hard part #2: numerical parity of factor statistics).

Oracle: the reference formulas written naively in fp64
(reference: kfac/utils.py:73-140)."""

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F
from hypothesis import given, settings, strategies as st

from kfac_pytorch_amd.ops.factors import (ComputeA, ComputeG,
                                          extract_patches, factor_dims,
                                          sym_factor)

computeA = ComputeA()
computeG = ComputeG()


def naive_linear_A(a, bias):
    a64 = a.double()
    if a64.dim() > 2:
        a64 = a64.mean(dim=tuple(range(1, a64.dim() - 1)))
    if bias:
        a64 = torch.cat([a64, torch.ones(a64.shape[0], 1,
                                         dtype=torch.float64)], 1)
    return a64.t() @ (a64 / a64.shape[0])


@settings(max_examples=40, deadline=None, derandomize=True)
@given(b=st.integers(1, 6), din=st.integers(1, 17),
       seq=st.integers(0, 5), bias=st.booleans(),
       seed=st.integers(0, 10 ** 6))
def test_linear_A_matches_reference_formula(b, din, seq, bias, seed):
    g = torch.Generator().manual_seed(seed)
    shape = (b, seq, din) if seq else (b, din)
    a = torch.randn(*shape, generator=g)
    layer = nn.Linear(din, 3, bias=bias)
    got = computeA(a, layer)
    ref = naive_linear_A(a, bias)
    assert got.shape == ref.shape
    torch.testing.assert_close(got.double(), ref, rtol=1e-4, atol=1e-6)


@settings(max_examples=40, deadline=None, derandomize=True)
@given(b=st.integers(1, 6), dout=st.integers(1, 17),
       seq=st.integers(0, 5), avg=st.booleans(),
       seed=st.integers(0, 10 ** 6))
def test_linear_G_matches_reference_formula(b, dout, seq, avg, seed):
    g = torch.Generator().manual_seed(seed)
    shape = (b, seq, dout) if seq else (b, dout)
    gr = torch.randn(*shape, generator=g)
    layer = nn.Linear(3, dout)
    got = computeG(gr, layer, batch_averaged=avg)
    g64 = gr.double()
    if g64.dim() > 2:
        g64 = g64.mean(dim=tuple(range(1, g64.dim() - 1)))
    B = g64.shape[0]
    # reference formula (kfac/utils.py:131-140): batch_averaged ->
    # G = g^T (g*B) = B * g^T g; else G = g^T (g/B)
    ref = B * (g64.t() @ g64) if avg else g64.t() @ (g64 / B)
    torch.testing.assert_close(got.double(), ref, rtol=1e-4, atol=1e-6)


@settings(max_examples=25, deadline=None, derandomize=True)
@given(b=st.integers(1, 3), cin=st.integers(1, 5), hw=st.integers(3, 8),
       k=st.integers(1, 3), stride=st.integers(1, 2),
       pad=st.integers(0, 1), bias=st.booleans(),
       seed=st.integers(0, 10 ** 6))
def test_conv_A_matches_unfold_reference(b, cin, hw, k, stride, pad,
                                         bias, seed):
    if hw + 2 * pad < k:
        return
    g = torch.Generator().manual_seed(seed)
    a = torch.randn(b, cin, hw, hw, generator=g)
    layer = nn.Conv2d(cin, 4, k, stride=stride, padding=pad, bias=bias)
    got = computeA(a, layer)
    # reference order (kfac/utils.py:86-94): unfold patches, append
    # ones BEFORE the spatial division, then a^T (a / B)
    patches = F.unfold(a.double(), k, padding=pad, stride=stride)
    patches = patches.transpose(1, 2).reshape(-1, patches.shape[1])
    spatial = patches.shape[0] // b
    if bias:
        patches = torch.cat(
            [patches, torch.ones(patches.shape[0], 1,
                                 dtype=torch.float64)], 1)
    patches = patches / spatial
    ref = patches.t() @ (patches / b)
    da, _ = factor_dims(layer)
    assert got.shape == (da, da)
    torch.testing.assert_close(got.double(), ref, rtol=1e-4, atol=1e-6)


@settings(max_examples=30, deadline=None, derandomize=True)
@given(rows=st.integers(1, 40), d=st.integers(1, 20),
       rs=st.floats(0.1, 4.0), denom=st.floats(0.5, 8.0),
       bias=st.booleans(), decay=st.floats(0.05, 0.99),
       seed=st.integers(0, 10 ** 6))
def test_sym_factor_running_average_property(rows, d, rs, denom, bias,
                                             decay, seed):
    """out' = (1-decay)*out + decay*F for ANY parameters, and F is
    symmetric PSD."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(rows, d, generator=g)
    F_ = sym_factor(x, row_scale=rs, denom=denom, bias=bias)
    torch.testing.assert_close(F_, F_.t(), rtol=1e-5, atol=1e-6)
    ev = torch.linalg.eigvalsh(F_.double())
    assert float(ev.min()) > -1e-6 * max(1.0, float(ev.max()))
    out = torch.randn(F_.shape[0], F_.shape[0], generator=g)
    expect = (1 - decay) * out + decay * F_
    got = sym_factor(x, row_scale=rs, denom=denom, bias=bias,
                     out=out.clone(), decay=decay)
    torch.testing.assert_close(got, expect, rtol=1e-4, atol=1e-5)


@settings(max_examples=20, deadline=None, derandomize=True)
@given(b=st.integers(1, 3), cin=st.integers(1, 4), hw=st.integers(2, 6),
       k=st.integers(1, 2), seed=st.integers(0, 10 ** 6))
def test_extract_patches_matches_unfold(b, cin, hw, k, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(b, cin, hw, hw, generator=g)
    got = extract_patches(x, (k, k), (1, 1), (0, 0), (1, 1))
    ref = F.unfold(x, k).transpose(1, 2).reshape(got.shape)
    torch.testing.assert_close(got, ref)
