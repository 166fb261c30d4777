"""GPU tests for the async/batched rocSOLVER tier
(``kfac_pytorch_amd.ops._kfac_solver``) and the linalg dispatch on top.

Numerics oracle: torch.linalg.eigh / cholesky_inverse in fp32
(same-matrix comparisons, reconstruction-based where eigenvector sign
and degenerate ordering make direct comparison ill-posed)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def spd(m, seed=0, device="cuda"):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(m, m, generator=g).to(device)
    return x @ x.t() / m + 0.1 * torch.eye(m, device=device)


@pytest.fixture(scope="module")
def solver():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from kfac_pytorch_amd.ops import _ext
    assert _ext.has_solver(), "_kfac_solver must be built on a GPU box"
    return _ext.load_solver()


def check_eig(A, w, Q, tol=5e-3):
    m = A.shape[-1]
    recon_err = ((Q @ torch.diag(w) @ Q.mT - A).norm() / A.norm()).item()
    orth_err = ((Q.mT @ Q - torch.eye(m, device=A.device)).norm()
                / (m ** 0.5)).item()
    assert recon_err < tol, f"reconstruction {recon_err:.2e}"
    assert orth_err < tol, f"orthogonality {orth_err:.2e}"


def test_syevdj_batched_matches_eigh(solver):
    mats = torch.stack([spd(192, seed=i) for i in range(5)])
    stacked = mats.clone()
    W, info = solver.syevdj_batched_(stacked)
    torch.cuda.synchronize()
    assert int(info.abs().sum()) == 0
    for k in range(5):
        check_eig(mats[k], W[k], stacked[k].mT)
        w_ref = torch.linalg.eigvalsh(mats[k])
        torch.testing.assert_close(W[k], w_ref, rtol=1e-3, atol=1e-3)


def test_syevd_pool_mixed_sizes(solver):
    sizes = [256, 513, 1024, 300, 700, 2048]
    mats = [spd(m, seed=m) for m in sizes]
    work = [a.clone() for a in mats]
    res = solver.syevd_pool_(work)
    torch.cuda.synchronize()
    info = res[-1]
    assert int(info.abs().sum()) == 0
    for k, a in enumerate(mats):
        check_eig(a, res[k], work[k].mT)


def test_potri_pool_matches_cholesky_inverse(solver):
    sizes = [128, 500, 1111]
    mats = [spd(m, seed=m + 7) for m in sizes]
    work = [a.clone() for a in mats]
    (info,) = solver.potri_pool_(work)
    torch.cuda.synchronize()
    assert int(info.abs().sum()) == 0
    for a, w in zip(mats, work):
        inv = w.triu(0) + w.triu(1).mT
        ref = torch.cholesky_inverse(torch.linalg.cholesky(a))
        err = ((inv - ref).norm() / ref.norm()).item()
        assert err < 1e-3, err


def test_mat_eig_multi_gpu_dispatch(solver):
    """Full dispatch: <=128 -> LDS-Jacobi kernel, same-dim groups ->
    syevdj batched, singles -> pool.  ResNet-50-like dim multiset."""
    from kfac_pytorch_amd.ops.linalg import mat_eig_multi
    dims = [64, 64, 128, 256, 256, 256, 576, 576, 1024, 2048, 2304, 4608]
    mats = [spd(m, seed=m + k) for k, m in enumerate(dims)]
    out = mat_eig_multi(mats, need_sorted=False)
    torch.cuda.synchronize()
    for a, (w, Q) in zip(mats, out):
        check_eig(a, w, Q)


def test_mat_inv_multi_gpu(solver):
    from kfac_pytorch_amd.ops.linalg import mat_inv_multi
    dims = [64, 300, 1025]
    mats = [spd(m, seed=m) for m in dims]
    damps = [0.1, 0.2, 0.3]
    out = mat_inv_multi(mats, damp_diag=damps)
    torch.cuda.synchronize()
    for a, d, inv in zip(mats, damps, out):
        damped = a + d * torch.eye(a.shape[0], device=a.device)
        ref = torch.linalg.inv(damped)
        err = ((inv - ref).norm() / ref.norm()).item()
        assert err < 1e-3, err


def test_eigen_dp_step_uses_solver_path(solver):
    """End-to-end eigen_dp step on a model with >128-dim factors (forces
    the rocSOLVER tier) -- gradients stay finite and preconditioned."""
    import torch.nn as nn
    import torch.nn.functional as F
    import torch.distributed as dist
    from tests.conftest import free_port
    import kfac_pytorch_amd as kfac
    import kfac_pytorch_amd.parallel.comm as comm_mod
    if dist.is_initialized():
        dist.destroy_process_group()
    comm_mod.reset()
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{free_port()}",
        world_size=1, rank=0)
    comm_mod.init("Torch")
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(512, 300), nn.ReLU(),
                          nn.Linear(300, 517), nn.ReLU(),
                          nn.Linear(517, 10)).cuda()
    pre = kfac.get_kfac_module("eigen_dp")(model, damping=0.003)
    x = torch.randn(8, 512, device="cuda")
    for _ in range(2):
        loss = F.cross_entropy(model(x), torch.randint(0, 10, (8,),
                                                       device="cuda"))
        loss.backward()
        pre.step()
    torch.cuda.synchronize()
    assert all(torch.isfinite(p.grad).all() for p in model.parameters())
    dist.destroy_process_group()
    comm_mod.reset()


def test_mat_eig_multi_padded_buckets(solver):
    """Nearby dims are padded into one batched syevd (pad block = -1
    diagonal): results must still match eigh exactly on the real part."""
    from kfac_pytorch_amd.ops.linalg import mat_eig_multi
    dims = [2048, 2049, 2304, 1000, 1024, 1152, 513, 512, 576]
    mats = [spd(m, seed=m) for m in dims]
    out = mat_eig_multi(mats, need_sorted=False)
    torch.cuda.synchronize()
    for a, (w, Q) in zip(mats, out):
        assert w.shape[0] == a.shape[0]
        assert Q.shape == a.shape
        check_eig(a, w, Q)
        w_ref = torch.linalg.eigvalsh(a)
        torch.testing.assert_close(torch.sort(w).values, w_ref,
                                   rtol=2e-3, atol=2e-3)


def test_mat_inv_multi_padded_batched(solver):
    """Bucketed-padding batched potrf+potri path vs torch inverse."""
    from kfac_pytorch_amd.ops.linalg import mat_inv_multi
    dims = [2048, 2049, 2304, 1000, 1024, 300]
    mats = [spd(m, seed=m + 3) for m in dims]
    damps = [0.05] * len(dims)
    out = mat_inv_multi(mats, damp_diag=damps)
    torch.cuda.synchronize()
    for a, d, inv in zip(mats, damps, out):
        damped = a + d * torch.eye(a.shape[0], device=a.device)
        ref = torch.linalg.inv(damped)
        err = ((inv - ref).norm() / ref.norm()).item()
        assert err < 1e-3, (a.shape, err)


def test_custom_sytrd_direct(solver):
    """Hand-written batched tridiagonalization vs torch.linalg.eigvalsh:
    the tridiagonal T must be orthogonally similar to A (eigenvalues
    match) and no matrix may abort."""
    if not hasattr(solver, "sytrd_batched_custom_"):
        pytest.skip("old _kfac_solver build")
    mats = torch.stack([spd(1152, seed=11 + i) for i in range(3)])
    work = mats.clone()
    E, tau, status = solver.sytrd_batched_custom_(work)
    torch.cuda.synchronize()
    assert int(status.abs().sum()) == 0, status.tolist()
    for k in range(mats.shape[0]):
        n = mats.shape[-1]
        D = work[k].diagonal()
        T = (torch.diag(D) + torch.diag(E[k][:n - 1], 1)
             + torch.diag(E[k][:n - 1], -1))
        ev = torch.linalg.eigvalsh(T)
        ev_ref = torch.linalg.eigvalsh(mats[k])
        scale = float(ev_ref.abs().max())
        err = float((ev - ev_ref).abs().max()) / scale
        assert err < 5e-5, (k, err)


def test_custom_sytrd_full_path(solver, monkeypatch):
    """Full custom tier through mat_eig_multi (sytrd -> stedc -> WY
    back-transform), forced on for every bucket size, with padded
    buckets and rank-deficient members -- vs the eigh oracle."""
    if not hasattr(solver, "sytrd_batched_custom_"):
        pytest.skip("old _kfac_solver build")
    from kfac_pytorch_amd.ops.linalg import mat_eig_multi
    monkeypatch.setenv("KFAC_SYTRD_MIN", "1000")
    monkeypatch.setenv("KFAC_CUSTOM_SYTRD", "1")
    # 2049/2304 pad into one bucket; 1397 tests the n%4 rounding;
    # the rank-deficient member mimics early-training conv factors
    dims = [2304, 2049, 2048, 1397, 1152, 1152]
    mats = [spd(m, seed=m) for m in dims]
    lowrank = torch.randn(1152, 200, device="cuda")
    mats[-1] = lowrank @ lowrank.t() / 200
    out = mat_eig_multi(mats, need_sorted=False)
    torch.cuda.synchronize()
    for a, (w, Q) in zip(mats, out):
        assert w.shape[0] == a.shape[0]
        assert Q.shape == a.shape
        check_eig(a, w, Q)
        w_ref = torch.linalg.eigvalsh(a)
        torch.testing.assert_close(torch.sort(w).values, w_ref,
                                   rtol=2e-3, atol=2e-3)


def test_custom_sytrd_precondition_accuracy(solver, monkeypatch):
    """What K-FAC actually consumes: the damped-inverse built from the
    custom path's eigenpairs must match the exact eigh's."""
    if not hasattr(solver, "sytrd_batched_custom_"):
        pytest.skip("old _kfac_solver build")
    from kfac_pytorch_amd.ops.linalg import mat_eig_multi
    monkeypatch.setenv("KFAC_SYTRD_MIN", "1000")
    monkeypatch.setenv("KFAC_CUSTOM_SYTRD", "1")
    a = spd(1536, seed=5)
    (res,) = [mat_eig_multi([a], need_sorted=False)[0]]
    torch.cuda.synchronize()
    w, Q = res
    damping = 0.002
    P = Q @ torch.diag(1.0 / (w.clamp_min(0) + damping)) @ Q.mT
    w_ref, Q_ref = torch.linalg.eigh(a)
    P_ref = (Q_ref @ torch.diag(1.0 / (w_ref.clamp_min(0) + damping))
             @ Q_ref.mT)
    err = ((P - P_ref).norm() / P_ref.norm()).item()
    assert err < 5e-3, err


def test_mat_inv_multi_trsm_tier(solver):
    """The GEMM-rate inverse tier (chol + trsm + X^T X) for buckets
    >= 512 must match torch.cholesky_inverse exactly, padded buckets
    and singletons included."""
    from kfac_pytorch_amd.ops.linalg import mat_inv_multi
    dims = [2304, 2049, 2048, 1152, 700, 512, 256, 256, 64]
    mats = [spd(m, seed=m + 3) for m in dims]
    out = mat_inv_multi(mats, damp_diag=[0.01] * len(mats))
    torch.cuda.synchronize()
    for a, inv in zip(mats, out):
        damped = a + 0.01 * torch.eye(a.shape[0], device="cuda")
        ref = torch.cholesky_inverse(torch.linalg.cholesky(damped))
        err = ((inv - ref).norm() / ref.norm()).item()
        assert err < 1e-3, (a.shape[0], err)


def test_sbr_stage1_gpu_similarity():
    """SBR stage-1 band reduction (ops/sbr.py) on hardware: the
    sync-free deferred path at a real bucket dim, fp32 -- similarity
    and orthogonality at the fp32 tolerance measured in
    profiles/sbr_stage1_deferred.log (4e-6 at 1152)."""
    from kfac_pytorch_amd.ops.sbr import (apply_q_batched,
                                          band_reduce_batched)
    g = torch.Generator().manual_seed(5)
    x = torch.randn(2, 576, 1152, generator=g)
    A = (x @ x.mT / 1152).cuda()
    B, panels = band_reduce_batched(A, 64, check="deferred")
    assert float(B.triu(65).abs().max()) == 0.0
    eye = torch.eye(576, device="cuda").expand(2, -1, -1).contiguous()
    Q = apply_q_batched(panels, eye)
    assert float((Q @ Q.mT - eye).abs().max()) < 2e-5
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 5e-5
