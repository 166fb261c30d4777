"""GPU numerics tests for the CDNA4 HIP kernels -- each kernel compared
against the plain PyTorch fp32 implementation of the same op.

All tests here require an MI355X (run via gpurun)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs GPU")


@pytest.fixture(scope="module")
def ext():
    from kfac_pytorch_amd.ops import _ext
    assert _ext.available(), "HIP extension must be built on the GPU box"
    return _ext


def cpu_sym_factor(x, row_scale, denom, bias):
    """fp32 torch oracle of the SYRK factor kernel's math."""
    x = x.float().cpu()
    if bias:
        x = torch.cat([x, x.new_ones(x.size(0), 1)], 1)
    return (x.t() @ x) * (row_scale * row_scale / denom)


@requires_gpu
@pytest.mark.parametrize("rows,d,bias", [
    (64, 16, False),        # single tile, tiny
    (64, 16, True),         # bias column
    (1000, 300, True),      # multi-tile (catches transposed C-writes)
    (1000, 300, False),
    (50000, 27, True),      # deep split-K, ResNet first-conv shape
    (4096, 513, True),      # fc-layer shape, odd dim
    (777, 129, True),       # nothing aligned
])
def test_syrk_factor_vs_oracle(ext, rows, d, bias):
    torch.manual_seed(rows + d)
    x = torch.randn(rows, d, device="cuda").bfloat16()
    n = d + (1 if bias else 0)
    out = torch.empty(n, n, device="cuda")
    ext.syrk_factor_(x, out, 0.37, float(rows), bias, -1.0)
    ref = cpu_sym_factor(x, 0.37, float(rows), bias)
    torch.testing.assert_close(out.cpu(), ref, rtol=3e-2, atol=3e-3)


@requires_gpu
def test_syrk_factor_decay_epilogue(ext):
    torch.manual_seed(0)
    rows, d = 512, 100
    x = torch.randn(rows, d, device="cuda").bfloat16()
    out = torch.eye(d + 1, device="cuda")
    prev = out.clone()
    ext.syrk_factor_(x, out, 1.0, float(rows), True, 0.95)
    fresh = cpu_sym_factor(x, 1.0, float(rows), True).cuda()
    expected = 0.05 * prev + 0.95 * fresh
    torch.testing.assert_close(out, expected, rtol=3e-2, atol=3e-3)


@requires_gpu
def test_syrk_output_symmetric(ext):
    torch.manual_seed(3)
    x = torch.randn(2048, 400, device="cuda").bfloat16()
    out = torch.empty(401, 401, device="cuda")
    ext.syrk_factor_(x, out, 1.0, 2048.0, True, -1.0)
    torch.testing.assert_close(out, out.t(), rtol=0, atol=0)


@requires_gpu
def test_eigen_scale(ext):
    torch.manual_seed(1)
    ng, na, damping = 257, 130, 0.003
    v = torch.randn(ng, na, device="cuda")
    dG = torch.rand(ng, device="cuda")
    dA = torch.rand(na, device="cuda")
    expected = v / (dG.unsqueeze(1) * dA.unsqueeze(0) + damping)
    ext.eigen_scale_(v, dG, dA, damping)
    torch.testing.assert_close(v, expected, rtol=1e-6, atol=1e-7)


@requires_gpu
@pytest.mark.parametrize("shape,k,s,p,dil", [
    ((2, 3, 16, 16), (3, 3), (1, 1), (1, 1), (1, 1)),
    ((2, 8, 14, 14), (3, 3), (2, 2), (1, 1), (1, 1)),
    ((1, 4, 9, 9), (5, 5), (1, 1), (2, 2), (1, 1)),
    ((2, 3, 20, 20), (7, 7), (2, 2), (3, 3), (1, 1)),
    ((1, 2, 12, 12), (3, 3), (1, 1), (1, 1), (2, 2)),
])
def test_im2col_vs_cpu(ext, shape, k, s, p, dil):
    torch.manual_seed(5)
    import torch.nn.functional as F
    x = torch.randn(*shape, device="cuda")
    ours = ext.im2col(x, k[0], k[1], s[0], s[1], p[0], p[1], dil[0], dil[1])
    cols = F.unfold(x.cpu(), kernel_size=k, stride=s, padding=p,
                    dilation=dil)
    ref = cols.transpose(1, 2).reshape(-1, cols.size(1))
    assert ours.dtype == torch.bfloat16
    torch.testing.assert_close(ours.float().cpu(), ref, rtol=1e-2,
                               atol=1e-2)


@requires_gpu
def test_compute_factors_gpu_vs_cpu():
    """ComputeA/ComputeG full pipeline on GPU (HIP im2col + MFMA SYRK)
    vs the CPU oracle fed bf16-quantized inputs (isolates KERNEL
    correctness from the intended bf16-capture quantization: on random
    uncorrelated data the near-cancelling off-diagonal covariance sums
    amplify input rounding, which is capture policy, not kernel error)."""
    import torch.nn as nn
    from kfac_pytorch_amd.ops.factors import ComputeA, ComputeG
    torch.manual_seed(7)
    conv = nn.Conv2d(8, 16, 3, padding=1).cuda()
    a = torch.randn(4, 8, 14, 14, device="cuda")
    g = torch.randn(4, 16, 14, 14, device="cuda")
    A_gpu = ComputeA()(a, conv)
    G_gpu = ComputeG()(g, conv, True)
    a_q = a.cpu().bfloat16().float()
    g_q = g.cpu().bfloat16().float()
    A_cpu = ComputeA()(a_q, conv.cpu())
    G_cpu = ComputeG()(g_q, conv.cpu(), True)
    torch.testing.assert_close(A_gpu.cpu(), A_cpu, rtol=2e-3, atol=1e-4)
    # G rows are scaled by B*spatial before the product; scale-relative
    # tolerance
    torch.testing.assert_close(G_gpu.cpu(), G_cpu, rtol=2e-3,
                               atol=1e-2)


@requires_gpu
def test_eigen_precondition_gpu():
    from kfac_pytorch_amd.ops.linalg import eigen_precondition
    torch.manual_seed(11)
    na, ng, damping = 65, 33, 0.01
    A = torch.randn(na, na)
    A = (A @ A.t() / na + torch.eye(na)).cuda()
    G = torch.randn(ng, ng)
    G = (G @ G.t() / ng + torch.eye(ng)).cuda()
    dA, QA = torch.linalg.eigh(A)
    dG, QG = torch.linalg.eigh(G)
    grad = torch.randn(ng, na, device="cuda")
    ours = eigen_precondition(QA.contiguous(), dA, QG.contiguous(), dG,
                              grad, damping)
    v1 = QG.t() @ grad @ QA
    v2 = v1 / (dG.unsqueeze(1) * dA.unsqueeze(0) + damping)
    ref = QG @ v2 @ QA.t()
    torch.testing.assert_close(ours, ref, rtol=1e-4, atol=1e-5)


@requires_gpu
def test_e2e_eigen_dp_step_gpu(single_process_comm):
    """End-to-end K-FAC step on GPU with the native kernels in the path."""
    import torch.nn.functional as F
    import kfac_pytorch_amd as kfac
    from kfac_pytorch_amd.models import get_cifar_model
    torch.manual_seed(13)
    model = get_cifar_model("resnet20").cuda()
    pre = kfac.get_kfac_module("eigen_dp")(model, lr=0.1, damping=0.003)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    for step in range(3):
        x = torch.randn(8, 3, 32, 32, device="cuda")
        y = torch.randint(0, 10, (8,), device="cuda")
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        pre.step()
        opt.step()
    assert all(torch.isfinite(p).all() for p in model.parameters())
    assert pre.steps == 3


@requires_gpu
def test_native_extension_is_loaded():
    from kfac_pytorch_amd.ops import _ext
    assert _ext.available()
    from kfac_pytorch_amd.ops import _kfac_hip
    assert "_kfac_hip" in _kfac_hip.__file__


@requires_gpu
@pytest.mark.parametrize("m", [4, 27, 64, 65, 100, 128])
def test_jacobi_eigh_vs_eigh(ext, m):
    torch.manual_seed(m)
    a = torch.randn(m, m)
    a = (a @ a.t() / m + 0.1 * torch.eye(m)).cuda()
    w, V = ext.jacobi_eigh(a)
    w_ref = torch.linalg.eigvalsh(a)
    torch.testing.assert_close(w, w_ref, rtol=1e-4, atol=1e-5)
    # reconstruction + orthogonality (eigenvectors are sign/degenerate
    # ambiguous, so compare via V diag(w) V^T)
    recon = V @ torch.diag(w) @ V.t()
    torch.testing.assert_close(recon, a, rtol=1e-4, atol=1e-4)
    eye = torch.eye(m, device="cuda")
    torch.testing.assert_close(V.t() @ V, eye, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_jacobi_eigh_batched_mixed_sizes(ext):
    torch.manual_seed(0)
    mats = []
    for m in (27, 65, 128, 10, 128):
        a = torch.randn(m, m)
        mats.append((a @ a.t() / m + 0.05 * torch.eye(m)).cuda())
    results = ext.jacobi_eigh_batched(mats)
    for a, (w, V) in zip(mats, results):
        recon = V @ torch.diag(w) @ V.t()
        torch.testing.assert_close(recon, a, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_mat_eig_auto_uses_jacobi(ext):
    from kfac_pytorch_amd.ops.linalg import mat_eig, mat_eig_multi
    torch.manual_seed(2)
    a = torch.randn(96, 96)
    a = (a @ a.t() / 96 + 0.1 * torch.eye(96)).cuda()
    d, Q = mat_eig(a, method="auto")
    d_ref, _ = torch.linalg.eigh(a)
    torch.testing.assert_close(d, d_ref, rtol=1e-4, atol=1e-5)
    big = torch.randn(300, 300)
    big = (big @ big.t() / 300 + 0.1 * torch.eye(300)).cuda()
    outs = mat_eig_multi([a, big], need_sorted=False)
    for src, (w, V) in zip([a, big], outs):
        recon = V @ torch.diag(w) @ V.t()
        torch.testing.assert_close(recon, src, rtol=1e-4, atol=1e-4)


def test_eigen_scale_batched(ext):
    torch.manual_seed(5)
    nb, ng, na = 4, 33, 57
    v = torch.randn(nb, ng, na, device="cuda")
    dG = torch.rand(nb, ng, device="cuda")
    dA = torch.rand(nb, na, device="cuda")
    ref = v / (dG.unsqueeze(2) * dA.unsqueeze(1) + 0.003)
    ext.eigen_scale_batched_(v, dG, dA, 0.003)
    torch.cuda.synchronize()
    torch.testing.assert_close(v, ref, rtol=1e-6, atol=1e-7)


def test_eigen_precondition_multi_matches_single():
    """Shape-grouped bmm path vs the per-layer oracle."""
    from kfac_pytorch_amd.ops.linalg import (eigen_precondition,
                                             eigen_precondition_multi)
    torch.manual_seed(6)
    shapes = [(64, 129), (64, 129), (64, 129), (128, 257), (32, 65)]
    QAs, dAs, QGs, dGs, grads = [], [], [], [], []
    for dg, da in shapes:
        QAs.append(torch.linalg.qr(
            torch.randn(da, da, device="cuda"))[0].contiguous())
        QGs.append(torch.linalg.qr(
            torch.randn(dg, dg, device="cuda"))[0].contiguous())
        dAs.append(torch.rand(da, device="cuda"))
        dGs.append(torch.rand(dg, device="cuda"))
        grads.append(torch.randn(dg, da, device="cuda"))
    outs = eigen_precondition_multi(QAs, dAs, QGs, dGs, grads, 0.002)
    for i in range(len(shapes)):
        ref = eigen_precondition(QAs[i], dAs[i], QGs[i], dGs[i],
                                 grads[i].clone(), 0.002)
        torch.testing.assert_close(outs[i], ref, rtol=1e-4, atol=1e-5)


def test_factor_overlap_matches_inline():
    """KFAC_FACTOR_OVERLAP side-stream factor computation must produce
    bitwise-compatible factors and preconditioned grads vs the inline
    phase (same kernels, different stream/timing)."""
    import os
    import torch.nn as nn
    import torch.nn.functional as F
    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.parallel.comm as comm_mod
    import torch.distributed as dist
    from tests.conftest import free_port
    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{free_port()}",
            world_size=1, rank=0)
    comm_mod.reset()
    comm_mod.init("Torch")

    def run(flag):
        os.environ["KFAC_FACTOR_OVERLAP"] = flag
        torch.manual_seed(3)
        model = nn.Sequential(
            nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(), nn.Flatten(),
            nn.Linear(8 * 8 * 8, 10)).cuda()
        pre = kfac.KFAC_EIGEN_DP(model, damping=0.01)
        x = torch.randn(4, 3, 8, 8, device="cuda")
        y = torch.randint(0, 10, (4,), device="cuda")
        for _ in range(3):
            model.zero_grad(set_to_none=False)
            F.cross_entropy(model(x), y).backward()
            pre.step()
        torch.cuda.synchronize()
        A = {i: pre.m_A[m].clone() for i, m in enumerate(pre.modules)}
        G = {i: pre.m_G[m].clone() for i, m in enumerate(pre.modules)}
        grads = [p.grad.clone() for p in model.parameters()]
        os.environ.pop("KFAC_FACTOR_OVERLAP", None)
        return A, G, grads

    A1, G1, g1 = run("1")
    A0, G0, g0 = run("0")
    for k in A1:
        torch.testing.assert_close(A1[k], A0[k], rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(G1[k], G0[k], rtol=1e-5, atol=1e-6)
    for a, b in zip(g1, g0):
        torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-5)


def test_pred_graph_matches_eager():
    """hipGraph-captured pred phase must match the eager phase across
    steps, damping changes (recapture) and both eigen families."""
    import os
    import torch.nn as nn
    import torch.nn.functional as F
    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.parallel.comm as comm_mod
    import torch.distributed as dist
    from tests.conftest import free_port
    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{free_port()}",
            world_size=1, rank=0)
    comm_mod.reset()
    comm_mod.init("Torch")

    def run(flag, name):
        os.environ["KFAC_PRED_GRAPH"] = flag
        torch.manual_seed(5)
        model = nn.Sequential(
            nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(), nn.Flatten(),
            nn.Linear(8 * 8 * 8, 10)).cuda()
        pre = kfac.get_kfac_module(name)(model, damping=0.01)
        x = torch.randn(4, 3, 8, 8, device="cuda")
        y = torch.randint(0, 10, (4,), device="cuda")
        for step in range(4):
            if step == 2:  # force a recapture mid-run
                pre.param_groups[0]["damping"] = 0.02
            model.zero_grad(set_to_none=False)
            F.cross_entropy(model(x), y).backward()
            pre.step()
        torch.cuda.synchronize()
        out = [p.grad.clone() for p in model.parameters()]
        os.environ.pop("KFAC_PRED_GRAPH", None)
        return out

    for name in ("eigen_dp", "eigen", "inverse", "inverse_dp"):
        g1 = run("1", name)
        g0 = run("0", name)
        for a, b in zip(g1, g0):
            torch.testing.assert_close(a, b, rtol=1e-4, atol=1e-5)


def test_grouped_conv_step_gpu_matches_cpu():
    """Grouped-conv block factors on the GPU path (bf16 im2col ->
    fp32 batched bmm) must match the CPU oracle path."""
    import os
    import torch.nn as nn
    import torch.nn.functional as F
    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.parallel.comm as comm_mod
    import torch.distributed as dist
    from tests.conftest import free_port
    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{free_port()}",
            world_size=1, rank=0)
    comm_mod.reset()
    comm_mod.init("Torch")

    def run(device):
        torch.manual_seed(11)
        model = nn.Sequential(
            nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
            nn.Conv2d(8, 16, 3, groups=4, padding=1), nn.ReLU(),
            nn.Flatten(), nn.Linear(16 * 8 * 8, 10)).to(device)
        pre = kfac.KFAC_EIGEN_DP(model, damping=0.01)
        g = torch.Generator().manual_seed(2)
        x = torch.randn(4, 3, 8, 8, generator=g).to(device)
        y = torch.randint(0, 10, (4,), generator=g).to(device)
        for _ in range(2):
            model.zero_grad(set_to_none=False)
            F.cross_entropy(model(x), y).backward()
            pre.step()
        if device != "cpu":
            torch.cuda.synchronize()
        return [p.grad.detach().cpu().clone()
                for p in model.parameters()]

    ggpu = run("cuda")
    gcpu = run("cpu")
    for a, b in zip(ggpu, gcpu):
        # bf16 capture on GPU vs fp32 on CPU: loose but meaningful
        torch.testing.assert_close(a, b, rtol=5e-2, atol=5e-3)
