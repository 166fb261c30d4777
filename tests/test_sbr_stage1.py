"""Batched SBR stage-1 band reduction (ops/sbr.py) against exact
similarity/orthogonality/spectrum oracles -- the torch counterpart of
the validated numpy reference (scripts/sbr_ref.py, tested in
tests/test_sytrd_ref.py)."""

import pytest
import torch

from kfac_pytorch_amd.ops.sbr import apply_q_batched, band_reduce_batched


def _spd_stack(N, n, dtype, seed=0, rank=None):
    g = torch.Generator().manual_seed(seed)
    r = rank or 2 * n
    x = torch.randn(N, n, r, generator=g, dtype=dtype)
    return x @ x.mT / r


@pytest.mark.parametrize("n,b", [(65, 8), (130, 16), (200, 32),
                                 (192, 64)])
def test_band_reduce_similarity_fp64(n, b):
    A = _spd_stack(3, n, torch.float64, seed=n)
    B, panels = band_reduce_batched(A, b)
    # band structure is exact by construction
    assert float(B.triu(b + 1).abs().max()) == 0.0
    assert torch.allclose(B, B.mT)
    # Q from the factored panels: orthogonal, and Q B Q^T == A
    eye = torch.eye(n, dtype=torch.float64).expand(3, n, n).contiguous()
    Q = apply_q_batched(panels, eye)
    assert float((Q @ Q.mT - eye).abs().max()) < 1e-12
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 1e-13
    # spectrum preserved
    ev = torch.linalg.eigvalsh(B)
    ev_ref = torch.linalg.eigvalsh(A)
    scale = float(ev_ref.abs().max())
    assert float((ev - ev_ref).abs().max()) / scale < 1e-12


def test_band_reduce_fp32_tolerance():
    """fp32 (the GPU compute dtype for K-FAC factors): similarity to
    ~1e-5 relative, matching the existing solver tiers' accuracy."""
    A = _spd_stack(2, 150, torch.float32, seed=7)
    B, panels = band_reduce_batched(A, 32)
    eye = torch.eye(150).expand(2, 150, 150).contiguous()
    Q = apply_q_batched(panels, eye)
    assert float((Q @ Q.mT - eye).abs().max()) < 5e-6
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 2e-5


def test_band_reduce_rank_deficient():
    """Early-training K-FAC factors are rank-deficient sample
    covariances: the tau=0 degenerate-reflector path must stay an
    exact similarity."""
    A = _spd_stack(2, 96, torch.float64, seed=3, rank=20)
    B, panels = band_reduce_batched(A, 16)
    eye = torch.eye(96, dtype=torch.float64).expand(2, 96, 96)
    Q = apply_q_batched(panels, eye.contiguous())
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 1e-12


def test_band_reduce_small_matrix_noop():
    """n <= b+1: nothing to annihilate, Q = I."""
    A = _spd_stack(1, 8, torch.float64)
    B, panels = band_reduce_batched(A, 8)
    assert panels == []
    assert torch.allclose(B, A)


def test_band_eigenvectors_back_transform():
    """End-to-end use shape: eigh of the (dense-stored) band matrix,
    back-transformed by apply_q -- must reproduce eigh(A)."""
    A = _spd_stack(2, 120, torch.float64, seed=11)
    B, panels = band_reduce_batched(A, 16)
    d, Z = torch.linalg.eigh(B)
    Qz = apply_q_batched(panels, Z)
    # A Qz == Qz diag(d)
    resid = (A @ Qz - Qz * d.unsqueeze(-2)).abs().max()
    assert float(resid) < 1e-11


def test_fast_path_taken_without_geqrf_fallback(monkeypatch):
    """Well-conditioned panels must run the batched CholeskyQR +
    Householder-reconstruction path -- the geqrf fallback is for
    rank-deficient/tail/pivoting panels only (the GPU perf story
    depends on this)."""
    import kfac_pytorch_amd.ops.sbr as sbr
    calls = {"n": 0}
    orig = sbr._panel_wy_geqrf

    def counting(P):
        calls["n"] += 1
        return orig(P)

    monkeypatch.setattr(sbr, "_panel_wy_geqrf", counting)
    A = _spd_stack(2, 200, torch.float64, seed=5)
    B, panels = band_reduce_batched(A, 32)
    # only the tail panels (M <= b) may fall back
    assert calls["n"] <= 2, calls["n"]
    eye = torch.eye(200, dtype=torch.float64).expand(2, -1, -1)
    Q = apply_q_batched(panels, eye.contiguous())
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 1e-13


def test_panel_wy_fast_matches_geqrf_transform():
    """The two panel factorizations differ in V/T/sign conventions but
    must produce the SAME orthogonal action: H^T P = [R; 0] with
    identical R up to row signs, and identical H^T C H."""
    import kfac_pytorch_amd.ops.sbr as sbr
    g = torch.Generator().manual_seed(9)
    P = torch.randn(2, 80, 16, generator=g, dtype=torch.float64)

    def dense_h(V, Tinv):
        T = torch.linalg.solve_triangular(
            Tinv, torch.eye(V.shape[-1], dtype=V.dtype).expand(
                V.shape[0], -1, -1).contiguous(), upper=True)
        eye = torch.eye(V.shape[1], dtype=V.dtype)
        return eye - V @ T @ V.mT

    Hf = dense_h(*sbr._panel_wy(P)[:2])
    Hg = dense_h(*sbr._panel_wy_geqrf(P)[:2])
    for H in (Hf, Hg):
        HtP = H.mT @ P
        assert float(HtP[:, 16:].abs().max()) < 1e-12
    # the R blocks agree up to per-row sign
    Rf = (Hf.mT @ P)[:, :16]
    Rg = (Hg.mT @ P)[:, :16]
    sf = torch.sign(Rf.diagonal(dim1=-2, dim2=-1))
    sg = torch.sign(Rg.diagonal(dim1=-2, dim2=-1))
    assert torch.allclose(sf.unsqueeze(-1) * Rf, sg.unsqueeze(-1) * Rg,
                          atol=1e-11)


@pytest.mark.parametrize("check", ["deferred", "eager", "geqrf"])
def test_check_modes_agree(check):
    """All three failure-handling modes must produce an exact
    similarity; deferred (the sync-free GPU default) must match the
    reference geqrf path's band matrix up to roundoff."""
    A = _spd_stack(2, 150, torch.float64, seed=21)
    B, panels = band_reduce_batched(A, 32, check=check)
    eye = torch.eye(150, dtype=torch.float64).expand(2, -1, -1)
    Q = apply_q_batched(panels, eye.contiguous())
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 1e-13
    ev = torch.linalg.eigvalsh(B)
    ev_ref = torch.linalg.eigvalsh(A)
    assert float((ev - ev_ref).abs().max() / ev_ref.abs().max()) < 1e-12


def test_deferred_redo_on_rank_deficient():
    """Deferred mode on a rank-deficient stack must detect the flagged
    panels and redo on the geqrf path -- the result is still an exact
    similarity (NaNs from the poisoned fast pass must not escape)."""
    A = _spd_stack(2, 96, torch.float64, seed=3, rank=10)
    B, panels = band_reduce_batched(A, 16, check="deferred")
    assert bool(B.isfinite().all())
    eye = torch.eye(96, dtype=torch.float64).expand(2, -1, -1)
    Q = apply_q_batched(panels, eye.contiguous())
    resid = (Q @ B @ Q.mT - A).norm() / A.norm()
    assert float(resid) < 1e-12


# ------------------------------------------------- stage 2 + end-to-end
@pytest.mark.parametrize("n,b", [(24, 4), (48, 8), (65, 8)])
def test_bulge_chase_batched_tridiagonalizes(n, b):
    """Stage-2 torch chase (the HIP kernel's correctness reference):
    exact similarity band -> tridiagonal on a batch."""
    from kfac_pytorch_amd.ops.sbr import (band_reduce_batched,
                                          bulge_chase_batched)
    A = _spd_stack(3, n, torch.float64, seed=n + 1)
    B, panels = band_reduce_batched(A, b)
    T, Q2 = bulge_chase_batched(B, b)
    assert float(T.triu(2).abs().max()) < 1e-12      # tridiagonal
    eye = torch.eye(n, dtype=torch.float64)
    assert float((Q2 @ Q2.mT - eye).abs().max()) < 1e-12
    resid = (Q2 @ T @ Q2.mT - B).norm() / B.norm()
    assert float(resid) < 1e-13


@pytest.mark.parametrize("n,b", [(24, 4), (48, 8), (65, 8)])
def test_sbr_eigh_end_to_end(n, b):
    """Complete two-stage eigensolve vs torch.linalg.eigh."""
    from kfac_pytorch_amd.ops.sbr import sbr_eigh_batched
    A = _spd_stack(3, n, torch.float64, seed=n + 2)
    d, V = sbr_eigh_batched(A, b)
    resid = (A @ V - V * d.unsqueeze(-2)).abs().max()
    assert float(resid) < 1e-12
    eye = torch.eye(n, dtype=torch.float64)
    assert float((V @ V.mT - eye).abs().max()) < 1e-12
    ev = torch.linalg.eigvalsh(A)
    assert float((d - ev).abs().max() / ev.abs().max()) < 1e-12


def test_sbr_eigh_rank_deficient_fp32():
    """The K-FAC regime: rank-deficient fp32 sample covariance through
    the whole two-stage pipeline (stage-1 geqrf redo path + chase)."""
    from kfac_pytorch_amd.ops.sbr import sbr_eigh_batched
    A = _spd_stack(2, 48, torch.float32, seed=6, rank=10)
    d, V = sbr_eigh_batched(A, 8)
    resid = (A @ V - V * d.unsqueeze(-2)).abs().max()
    assert float(resid) < 1e-4
    ev = torch.linalg.eigvalsh(A)
    assert float((d - ev).abs().max() / ev.abs().max()) < 1e-4


@pytest.mark.parametrize("n,b", [(24, 4), (48, 8), (65, 8), (40, 16)])
def test_blocked_chase_matches_givens_oracle(n, b):
    """The reflector-blocked chase (the HIP kernel's exact per-hop
    math) against the per-element Givens oracle: both fix e1, so the
    implicit-Q theorem forces the same tridiagonal up to signs."""
    from kfac_pytorch_amd.ops.sbr import (band_reduce_batched,
                                          bulge_chase_batched,
                                          bulge_chase_blocked_batched)
    A = _spd_stack(3, n, torch.float64, seed=n + 3)
    B, _ = band_reduce_batched(A, b)
    T1, Q1 = bulge_chase_blocked_batched(B, b)
    assert float(T1.triu(2).abs().max()) == 0.0
    eye = torch.eye(n, dtype=torch.float64)
    assert float((Q1 @ Q1.mT - eye).abs().max()) < 1e-12
    resid = (Q1 @ T1 @ Q1.mT - B).norm() / B.norm()
    assert float(resid) < 1e-13
    T2, _ = bulge_chase_batched(B, b)
    dd = (T1.diagonal(dim1=-2, dim2=-1)
          - T2.diagonal(dim1=-2, dim2=-1)).abs().max()
    od = (T1.diagonal(offset=1, dim1=-2, dim2=-1).abs()
          - T2.diagonal(offset=1, dim1=-2, dim2=-1).abs()).abs().max()
    assert float(dd) < 1e-10 and float(od) < 1e-10


@pytest.mark.parametrize("chase", ["blocked", "givens"])
def test_sbr_eigh_both_chases(chase):
    from kfac_pytorch_amd.ops.sbr import sbr_eigh_batched
    A = _spd_stack(2, 48, torch.float64, seed=8)
    d, V = sbr_eigh_batched(A, 8, chase=chase)
    resid = (A @ V - V * d.unsqueeze(-2)).abs().max()
    assert float(resid) < 1e-12
    ev = torch.linalg.eigvalsh(A)
    assert float((d - ev).abs().max() / ev.abs().max()) < 1e-12


def test_wavefront_interleaving_commutes():
    """The HIP kernel's pipelining precondition: sweeps whose hop
    indices stay >= 2 apart act on disjoint row windows, so the
    interleaved (wavefront) schedule must produce the same
    tridiagonal as the sequential chase (exactly, in exact
    arithmetic; to roundoff here)."""
    import kfac_pytorch_amd.ops.sbr as sbr
    n, b = 48, 8
    A0 = _spd_stack(2, n, torch.float64, seed=31)
    B, _ = sbr.band_reduce_batched(A0, b)

    T_seq, Q_seq = sbr.bulge_chase_blocked_batched(B, b)

    A = B.clone()
    Q2 = torch.eye(n, dtype=A.dtype).expand(2, n, n).contiguous()
    gens, hops, nxt = {}, {}, 0
    while gens or nxt < n - 2:
        if nxt < n - 2 and (nxt == 0 or hops.get(nxt - 1, 0) >= 2
                            or (nxt - 1) not in gens):
            gens[nxt] = sbr._sweep_hops(A, Q2, nxt, b)
            hops[nxt] = 0
            nxt += 1
        for j in sorted(gens):
            try:
                next(gens[j])
                hops[j] += 1
            except StopIteration:
                del gens[j]
    T_wave = 0.5 * (A + A.mT)

    assert float(T_wave.triu(2).abs().max()) < 1e-12
    torch.testing.assert_close(T_wave, T_seq, rtol=1e-9, atol=1e-9)
    torch.testing.assert_close(Q2, Q_seq, rtol=1e-9, atol=1e-9)
