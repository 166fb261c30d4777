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
