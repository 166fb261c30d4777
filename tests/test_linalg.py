import pytest
import torch

from kfac_pytorch_amd.ops.linalg import (add_diagonal_, eigen_precondition,
                                         inverse_precondition, mat_eig,
                                         mat_inv)


def spd(n, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, n, generator=g)
    return x @ x.t() / n + 0.5 * torch.eye(n)


def test_add_diagonal_inplace():
    x = torch.zeros(4, 4)
    add_diagonal_(x, 2.5)
    torch.testing.assert_close(x, 2.5 * torch.eye(4))


@pytest.mark.parametrize("n", [1, 5, 64, 129])
def test_mat_inv_cholesky(n):
    a = spd(n, seed=n)
    inv = mat_inv(a)
    torch.testing.assert_close(inv @ a, torch.eye(n), rtol=1e-3, atol=1e-3)


def test_mat_inv_method_inv():
    a = spd(16)
    torch.testing.assert_close(mat_inv(a, "inv"), torch.linalg.inv(a),
                               rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("n", [2, 17, 128])
def test_mat_eig_reconstructs(n):
    a = spd(n, seed=n + 100)
    d, q = mat_eig(a, method="eigh")
    torch.testing.assert_close(q @ torch.diag(d) @ q.t(), a,
                               rtol=1e-4, atol=1e-4)
    # Q orthogonal
    torch.testing.assert_close(q.t() @ q, torch.eye(n), rtol=1e-4, atol=1e-4)


def test_eigen_precondition_matches_explicit_inverse():
    """Implicit-eigen preconditioning equals multiplying by the inverse of
    the damped Kronecker product: (G (x) A + damping I)^-1 vec(grad)."""
    na, ng, damping = 7, 5, 0.03
    A, G = spd(na, 1), spd(ng, 2)
    dA, QA = torch.linalg.eigh(A)
    dG, QG = torch.linalg.eigh(G)
    grad = torch.randn(ng, na)

    ours = eigen_precondition(QA.contiguous(), dA, QG.contiguous(), dG,
                              grad, damping)

    K = torch.kron(G, A) + damping * torch.eye(na * ng)
    # vec with row-major grad: vec(grad^T) ordering -> use kron(G, A) on
    # grad flattened as [ng*na] with G indexing rows
    expected = torch.linalg.solve(K, grad.reshape(-1)).reshape(ng, na)
    torch.testing.assert_close(ours, expected, rtol=1e-3, atol=1e-4)


def test_inverse_precondition():
    A, G = spd(4, 3), spd(6, 4)
    grad = torch.randn(6, 4)
    out = inverse_precondition(torch.linalg.inv(A), torch.linalg.inv(G), grad)
    expected = torch.linalg.inv(G) @ grad @ torch.linalg.inv(A)
    torch.testing.assert_close(out, expected, rtol=1e-4, atol=1e-5)


def test_pad_buckets_grouping():
    """Bucketed-padding grouping: members within PAD_RATIO of the
    leader share a bucket (ResNet-50's 2048/2049/2304 and
    1000/1024/1152 must merge)."""
    from kfac_pytorch_amd.ops.linalg import _pad_buckets
    dims = sorted([(4608, 0), (2304, 1), (2049, 2), (2048, 3),
                   (1152, 4), (1024, 5), (1000, 6), (576, 7), (512, 8)],
                  reverse=True)
    buckets = _pad_buckets(dims)
    as_sets = [(lead, sorted(i for _, i in members))
               for lead, members in buckets]
    assert as_sets[0] == (4608, [0])
    assert as_sets[1] == (2304, [1, 2, 3])
    assert as_sets[2] == (1152, [4, 5, 6])
    assert as_sets[3] == (576, [7, 8])


def test_eigen_precondition_equals_dense_kronecker_solve():
    """The implicit-eigen chain QG^T g QA / (dG dA^T + l) -> QG V QA^T
    must equal solving the dense damped Kronecker system
    (G (x) A + l I) vec(P) = vec(grad) directly -- the core K-FAC
    identity the whole eigen family rests on."""
    torch.manual_seed(4)
    da, dg, lam = 7, 5, 0.013
    xa = torch.randn(da, da, dtype=torch.float64)
    xg = torch.randn(dg, dg, dtype=torch.float64)
    A = xa @ xa.T / da + 0.1 * torch.eye(da, dtype=torch.float64)
    G = xg @ xg.T / dg + 0.1 * torch.eye(dg, dtype=torch.float64)
    grad = torch.randn(dg, da, dtype=torch.float64)

    dA, QA = torch.linalg.eigh(A)
    dG, QG = torch.linalg.eigh(G)
    P = eigen_precondition(QA, dA, QG, dG, grad.clone(), lam)

    # dense oracle: row-major vec with K = kron(G, A) matching
    # vec(G P A) = (G kron A^T)... use the transpose identity:
    # G P A = reshape((G kron A^T) vec_r(P)); A symmetric -> A^T = A
    K = torch.kron(G, A) + lam * torch.eye(dg * da, dtype=torch.float64)
    P_ref = torch.linalg.solve(K, grad.reshape(-1)).reshape(dg, da)

    torch.testing.assert_close(P, P_ref, rtol=1e-8, atol=1e-10)


def test_eigen_clamp_matches_reference_formula():
    """Eigenvalue clamp d*(d>eps) (reference
    kfac_preconditioner_base.py:115 / eigen.py:114,119) zeroes tiny and
    negative eigenvalues before the denominator."""
    d = torch.tensor([-1e-3, 0.0, 1e-12, 1e-9, 0.5])
    eps = 1e-10
    clamped = d * (d > eps)
    assert torch.equal(clamped,
                       torch.tensor([0.0, 0.0, 0.0, 1e-9, 0.5]))


def test_deferred_info_raises_once():
    """Nonzero rocSOLVER info words accumulate device-side and raise at
    the next check, then clear (the sync-free validation path)."""
    import pytest
    from kfac_pytorch_amd.ops.linalg import (_defer_info,
                                             check_deferred_info)
    check_deferred_info()  # clean state
    _defer_info(torch.tensor([0, 0, 0], dtype=torch.int32))
    check_deferred_info()  # zeros: no raise
    _defer_info(torch.tensor([0, 3, 0], dtype=torch.int32))
    _defer_info(torch.tensor([1], dtype=torch.int32))
    with pytest.raises(RuntimeError, match="2 rocSOLVER"):
        check_deferred_info()
    check_deferred_info()  # flag cleared by the raise
