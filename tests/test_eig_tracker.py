"""EigenTracker: warm-started block-Jacobi eigendecomposition tracking
(CPU path; the GPU path swaps in the batched LDS-Jacobi kernel)."""

import pytest
import torch

from kfac_pytorch_amd.ops.eig_tracker import EigenTracker


def slowly_varying_factors(m, steps, decay=0.95, seed=0):
    g = torch.Generator().manual_seed(seed)
    w = torch.randn(m, m, generator=g)
    base = w @ w.t() / m + 0.1 * torch.eye(m)
    A = torch.eye(m)
    for _ in range(steps):
        n = torch.randn(m, m, generator=g) / m
        fresh = base + (n + n.t()) / 2 * 0.2
        A = decay * A + (1 - decay) * fresh
        yield A.clone()


@pytest.mark.parametrize("m", [64, 200, 300])
def test_tracker_follows_slowly_varying_matrix(m):
    tracker = EigenTracker(cold_every=1000, rounds=3)
    colds = 0
    for t, A in enumerate(slowly_varying_factors(m, 50)):
        before = tracker.calls_since_cold
        w, Q = tracker.update(A)
        if tracker.calls_since_cold <= before and t > 0:
            colds += 1
        recon = Q @ torch.diag(w) @ Q.t()
        rel = (recon - A).norm() / A.norm()
        assert rel < 5e-3, f"step {t}: reconstruction error {rel:.2e}"
        orth = (Q.t() @ Q - torch.eye(m)).norm()
        assert orth < 1e-3, f"step {t}: orthogonality {orth:.2e}"
    # tracking should mostly stay warm
    assert colds <= 5


def test_tracker_cold_restart_on_jump():
    """A discontinuous jump in the matrix must trigger a cold restart
    (or at minimum keep the reconstruction accurate)."""
    g = torch.Generator().manual_seed(3)
    m = 150
    w1 = torch.randn(m, m, generator=g)
    A1 = w1 @ w1.t() / m + 0.1 * torch.eye(m)
    w2 = torch.randn(m, m, generator=g)
    A2 = w2 @ w2.t() / m + 0.1 * torch.eye(m)
    tracker = EigenTracker(cold_every=1000)
    tracker.update(A1)
    w, Q = tracker.update(A2)  # unrelated matrix
    recon = Q @ torch.diag(w) @ Q.t()
    assert (recon - A2).norm() / A2.norm() < 1e-4  # cold path accuracy


def test_tracker_cold_every_forces_restart():
    tracker = EigenTracker(cold_every=3)
    for t, A in enumerate(slowly_varying_factors(64, 8, seed=5)):
        tracker.update(A)
    assert tracker.calls_since_cold <= 3


def test_tracker_eigenvalues_match_eigh():
    tracker = EigenTracker(cold_every=1000, rounds=3)
    last = None
    for A in slowly_varying_factors(96, 20, seed=9):
        w, Q = tracker.update(A)
        last = (A, w)
    A, w = last
    w_ref = torch.linalg.eigvalsh(A)
    torch.testing.assert_close(torch.sort(w).values, w_ref,
                               rtol=1e-3, atol=1e-3)
