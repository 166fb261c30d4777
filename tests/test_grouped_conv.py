"""Grouped-convolution K-FAC: exact block-diagonal factors.

The gold oracle: a Conv2d(groups=g) is functionally identical to g
parallel ungrouped convs on channel slices with concatenated outputs.
Running the same K-FAC variant on both models (same weights, same
data) must produce identical preconditioned gradients -- this
exercises hooks, grouped factor products, per-block eigensolves /
Cholesky inverses, the batched pred path and the grad update.

(The reference computes a single dense -- and WRONG -- factor for
groups > 1; this framework preconditions each block exactly.)
"""

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from kfac_pytorch_amd.ops.factors import (ComputeA, ComputeG,
                                          factor_dims, factor_groups,
                                          sym_factor, sym_factor_grouped)


@pytest.fixture()
def single_comm():
    import torch.distributed as dist
    import kfac_pytorch_amd.parallel.comm as comm_mod
    from tests.conftest import free_port
    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{free_port()}",
            world_size=1, rank=0)
    comm_mod.reset()
    comm_mod.init("Torch")
    yield comm_mod.get_comm()


def test_sym_factor_grouped_matches_slices():
    torch.manual_seed(0)
    rows, g, d0 = 40, 4, 6
    x = torch.randn(rows, g * d0)
    for bias in (False, True):
        stacked = sym_factor_grouped(x, g, row_scale=0.5, denom=8.0,
                                     bias=bias)
        for i in range(g):
            ref = sym_factor(x[:, i * d0:(i + 1) * d0], row_scale=0.5,
                             denom=8.0, bias=bias)
            torch.testing.assert_close(stacked[i], ref, rtol=1e-5,
                                       atol=1e-6)


def test_grouped_factor_dims():
    m = nn.Conv2d(8, 12, 3, groups=4, bias=True)
    assert factor_groups(m) == 4
    da, dg = factor_dims(m)
    assert da == 2 * 9 + 1 and dg == 3


def test_grouped_compute_matches_per_group_layers():
    torch.manual_seed(1)
    gconv = nn.Conv2d(8, 12, 3, groups=4, padding=1, bias=True)
    a = torch.randn(5, 8, 6, 6)
    g_out = torch.randn(5, 12, 6, 6)
    A = ComputeA()(a, gconv)
    G = ComputeG()(g_out, gconv, batch_averaged=True)
    assert A.shape == (4, 19, 19) and G.shape == (4, 3, 3)
    for i in range(4):
        sub = nn.Conv2d(2, 3, 3, padding=1, bias=True)
        Ai = ComputeA()(a[:, 2 * i:2 * i + 2], sub)
        Gi = ComputeG()(g_out[:, 3 * i:3 * i + 3], sub,
                        batch_averaged=True)
        torch.testing.assert_close(A[i], Ai, rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(G[i], Gi, rtol=1e-5, atol=1e-6)


class _GroupedNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.pre = nn.Conv2d(3, 8, 3, padding=1)
        self.gc = nn.Conv2d(8, 12, 3, groups=4, padding=1, bias=True)
        self.head = nn.Linear(12 * 4 * 4, 5)

    def forward(self, x):
        x = F.relu(self.pre(x))
        x = F.relu(self.gc(x))
        return self.head(x.flatten(1))


class _PerGroupNet(nn.Module):
    """Functionally identical: 4 parallel ungrouped convs."""

    def __init__(self, src: _GroupedNet):
        super().__init__()
        self.pre = nn.Conv2d(3, 8, 3, padding=1)
        self.pre.load_state_dict(src.pre.state_dict())
        self.convs = nn.ModuleList(
            [nn.Conv2d(2, 3, 3, padding=1, bias=True) for _ in range(4)])
        with torch.no_grad():
            for i, c in enumerate(self.convs):
                c.weight.copy_(src.gc.weight[3 * i:3 * i + 3])
                c.bias.copy_(src.gc.bias[3 * i:3 * i + 3])
        self.head = nn.Linear(12 * 4 * 4, 5)
        self.head.load_state_dict(src.head.state_dict())

    def forward(self, x):
        x = F.relu(self.pre(x))
        xs = x.chunk(4, dim=1)
        x = torch.cat([c(xi) for c, xi in zip(self.convs, xs)], 1)
        x = F.relu(x)
        return self.head(x.flatten(1))


@pytest.mark.parametrize("name", ["eigen_dp", "inverse_dp", "eigen",
                                  "inverse"])
def test_grouped_step_matches_per_group_model(single_comm, name):
    import kfac_pytorch_amd as kfac
    torch.manual_seed(3)
    gm = _GroupedNet()
    pm = _PerGroupNet(gm)
    x = torch.randn(6, 3, 4, 4)
    y = torch.randint(0, 5, (6,))

    KFAC = kfac.get_kfac_module(name)
    pre_g = KFAC(gm, damping=0.01)
    pre_p = KFAC(pm, damping=0.01)
    for step in range(2):
        for model, pre in ((gm, pre_g), (pm, pre_p)):
            model.zero_grad(set_to_none=False)
            F.cross_entropy(model(x), y).backward()
            pre.step()
        # preconditioned grads of the grouped conv == the slices
        wg = gm.gc.weight.grad
        bg = gm.gc.bias.grad
        for i in range(4):
            torch.testing.assert_close(
                wg[3 * i:3 * i + 3], pm.convs[i].weight.grad,
                rtol=2e-4, atol=5e-6)
            torch.testing.assert_close(
                bg[3 * i:3 * i + 3], pm.convs[i].bias.grad,
                rtol=2e-4, atol=5e-6)
        # the shared layers must match too (same kl-clip nu)
        torch.testing.assert_close(gm.pre.weight.grad,
                                   pm.pre.weight.grad,
                                   rtol=2e-4, atol=5e-6)
        torch.testing.assert_close(gm.head.weight.grad,
                                   pm.head.weight.grad,
                                   rtol=2e-4, atol=5e-6)


def test_mobilenet_depthwise_now_preconditioned(single_comm):
    """MobileNetV2's depthwise convs (the round-1 coverage gap) are now
    hooked and get block factors."""
    import kfac_pytorch_amd as kfac
    from kfac_pytorch_amd.models import imagenet_extras as ex
    torch.manual_seed(0)
    model = ex.mobilenet_v2(num_classes=10)
    pre = kfac.KFAC_EIGEN_DP(model, damping=0.01)
    depthwise = [m for m in pre.modules
                 if isinstance(m, nn.Conv2d) and m.groups > 1]
    assert depthwise, "depthwise convs must be hooked now"
    x = torch.randn(2, 3, 64, 64)
    y = torch.randint(0, 10, (2,))
    model.zero_grad(set_to_none=False)
    F.cross_entropy(model(x), y).backward()
    pre.step()
    m0 = depthwise[0]
    assert pre.m_A[m0].dim() == 3
    assert torch.isfinite(m0.weight.grad).all()


def test_grouped_state_checkpoints(single_comm):
    """kfac_state_dict round-trips the 3-D grouped factor state."""
    import kfac_pytorch_amd as kfac
    torch.manual_seed(5)
    gm = _GroupedNet()
    pre = kfac.KFAC_EIGEN_DP(gm, damping=0.01)
    x = torch.randn(4, 3, 4, 4)
    y = torch.randint(0, 5, (4,))
    gm.zero_grad(set_to_none=False)
    F.cross_entropy(gm(x), y).backward()
    pre.step()
    from kfac_pytorch_amd.preconditioner.base import (
        kfac_state_dict, load_kfac_state_dict)
    sd = kfac_state_dict(pre)
    gm2 = _GroupedNet()
    pre2 = kfac.KFAC_EIGEN_DP(gm2, damping=0.01)
    gm2.zero_grad(set_to_none=False)
    F.cross_entropy(gm2(x), y).backward()
    pre2.step()  # allocate state
    load_kfac_state_dict(pre2, sd)
    torch.testing.assert_close(pre2.m_A[gm2.gc], pre.m_A[gm.gc])
    assert pre2.m_A[gm2.gc].dim() == 3


# ------------------------------------------------------------- distributed
def _worker_grouped_world2(rank, world, tmpfile, name):
    """Grouped-conv model under a real multi-rank step: owner-computed
    per-group factor blocks must broadcast/precondition identically on
    every rank (grouped bucket shapes are (g, d, d) -- the 3-D case of
    the owner-bucket comm path)."""
    import torch.distributed as dist
    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.parallel.comm as comm_mod
    dist.init_process_group("gloo", init_method=f"file://{tmpfile}",
                            world_size=world, rank=rank)
    comm_mod.reset()
    comm = comm_mod.init("Torch")
    torch.manual_seed(17)

    model = nn.Sequential(
        nn.Conv2d(4, 8, 3, padding=1), nn.ReLU(),
        nn.Conv2d(8, 8, 3, padding=1, groups=8, bias=True), nn.ReLU(),
        nn.Conv2d(8, 12, 3, padding=1, groups=2), nn.ReLU(),
        nn.Flatten(), nn.Linear(12 * 6 * 6, 5),
    )
    for p in model.parameters():
        comm.broadcast(p.data, src=0)
    pre = kfac.get_kfac_module(name)(model, damping=0.01)
    g = torch.Generator().manual_seed(100 + rank)
    x = torch.randn(6, 4, 6, 6, generator=g)
    y = torch.randint(0, 5, (6,), generator=g)
    for _ in range(2):
        model.zero_grad()
        F.cross_entropy(model(x), y).backward()
        for p in model.parameters():
            comm.allreduce(p.grad.data, op=comm.Average)
        pre.step()
        for p in model.parameters():
            mine = p.grad.clone()
            comm.broadcast(p.grad.data, src=0)
            torch.testing.assert_close(mine, p.grad, rtol=1e-4,
                                       atol=1e-5)
    dist.destroy_process_group()


@pytest.mark.parametrize("name", ["eigen", "eigen_dp", "inverse",
                                  "inverse_dp"])
def test_grouped_conv_world2_consistency(name):
    import os
    import tempfile
    from tests.conftest import spawn_retry

    def tmpfile():
        with tempfile.NamedTemporaryFile(delete=False) as f:
            name_ = f.name
        os.unlink(name_)
        return name_

    spawn_retry(_worker_grouped_world2, lambda: (2, tmpfile(), name), 2)
