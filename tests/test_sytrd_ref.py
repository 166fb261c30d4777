"""CPU oracle for the custom tridiagonalization algorithm
(scripts/sytrd_ref.py -- the exact deferred-alpha panel algebra the
HIP kernel ops/csrc_solver/sytrd_panel.hip implements), validated
against scipy/LAPACK-grade references in float64."""

import numpy as np
import pytest

from scripts.sytrd_ref import build_q, sytrd_blocked, tridiag


@pytest.mark.parametrize("n,nb", [(16, 8), (65, 16), (130, 64),
                                  (192, 64), (200, 32)])
def test_blocked_sytrd_matches_eigh(n, nb):
    rng = np.random.default_rng(n + nb)
    x = rng.standard_normal((n, 2 * n))
    a = x @ x.T / (2 * n)
    aout, d, e, tau = sytrd_blocked(a, nb)
    q = build_q(aout, tau)
    t = tridiag(d, e)
    assert np.linalg.norm(q @ t @ q.T - a) / np.linalg.norm(a) < 1e-12
    assert np.linalg.norm(q @ q.T - np.eye(n)) < 1e-11
    ev = np.linalg.eigvalsh(t)
    ev_ref = np.linalg.eigvalsh(a)
    scale = max(1e-12, np.abs(ev_ref).max())
    assert np.max(np.abs(ev - ev_ref)) / scale < 1e-11


def test_blocked_sytrd_rank_deficient():
    """K-FAC's big conv factors are rank-deficient sample covariances;
    the reduction must stay exact there (degenerate tau=0 columns)."""
    rng = np.random.default_rng(3)
    n, r = 96, 20
    x = rng.standard_normal((n, r))
    a = x @ x.T / r
    aout, d, e, tau = sytrd_blocked(a, 32)
    q = build_q(aout, tau)
    t = tridiag(d, e)
    assert np.linalg.norm(q @ t @ q.T - a) / np.linalg.norm(a) < 1e-12


def test_blocked_sytrd_diagonal_input():
    """An already-tridiagonal (diagonal) input exercises the tau=0
    branch in every column."""
    d_in = np.arange(1.0, 33.0)
    a = np.diag(d_in)
    aout, d, e, tau = sytrd_blocked(a, 8)
    assert np.allclose(d, d_in)
    assert np.allclose(e[:31], 0.0)
    assert np.allclose(tau[:31], 0.0)


@pytest.mark.parametrize("n,b", [(24, 4), (48, 8), (65, 8)])
def test_two_stage_sbr_reference(n, b):
    """Round-3 target algebra (docs/SBR_STAGE2_NOTES.md): full->band
    via CholeskyQR2 panels, band->tridiagonal via Givens bulge chasing
    -- both stages exact similarity transforms with the intended
    structure."""
    from scripts.sbr_ref import self_check
    r = self_check(n, b, seed=n)
    assert r["band_resid"] < 1e-12
    assert r["tri_resid"] < 1e-12
    assert r["recon_tri"] < 1e-12
    assert r["eig_err"] < 1e-12


def test_sbr_rank_deficient_panels():
    """CholeskyQR2's Householder fallback on rank-deficient panels
    (early-training K-FAC factors are rank-deficient)."""
    from scripts.sbr_ref import self_check
    import numpy as np
    rng = np.random.default_rng(1)
    n, r = 40, 8
    x = rng.standard_normal((n, r))
    a = x @ x.T / r          # rank 8 of 40
    from scripts.sbr_ref import band_reduce, bulge_chase
    B, Q1 = band_reduce(a, 4)
    T, Q2 = bulge_chase(B, 4)
    Q = Q1 @ Q2
    assert np.linalg.norm(Q @ T @ Q.T - a) / np.linalg.norm(a) < 1e-11
    assert np.max(np.abs(np.triu(T, 2))) < 1e-11
