"""Graph-support builders vs per-matrix reference math (GCN.py:49-138 semantics)."""

import pytest
import torch

from mpgcn_amd.graph import build_supports, get_support_K


def _ref_rw_normalize(A):
    d = A.sum(1)
    d_inv = torch.where(d == 0, torch.zeros_like(d), 1.0 / d)
    return torch.diag(d_inv) @ A


def _ref_cheb(x, order):
    T = [torch.eye(x.shape[0])]
    if order >= 1:
        T.append(x)
    for k in range(2, order + 1):
        T.append(2 * x @ T[k - 1] - T[k - 2])
    return torch.stack(T[: order + 1])


def test_support_K_contract():
    # Model_Trainer.py:24-36
    assert get_support_K("localpool", 1) == 1
    assert get_support_K("chebyshev", 2) == 3
    assert get_support_K("random_walk_diffusion", 2) == 3
    assert get_support_K("dual_random_walk_diffusion", 2) == 5
    with pytest.raises(ValueError):
        get_support_K("localpool", 2)
    with pytest.raises(ValueError):
        get_support_K("bogus", 2)


@pytest.mark.parametrize("kernel,order", [
    ("localpool", 1),
    ("chebyshev", 2),
    ("random_walk_diffusion", 2),
    ("random_walk_diffusion", 3),
    ("dual_random_walk_diffusion", 2),
])
def test_shapes_and_batching(kernel, order):
    torch.manual_seed(0)
    B, N = 5, 23
    flow = torch.rand(B, N, N) * 10
    out = build_supports(flow, kernel, order)
    K = get_support_K(kernel, order)
    assert out.shape == (B, K, N, N)
    # batched result == per-sample result
    for b in range(B):
        single = build_supports(flow[b:b + 1], kernel, order)[0]
        assert torch.allclose(out[b], single, atol=1e-5)


def test_random_walk_diffusion_matches_reference_math():
    torch.manual_seed(1)
    N = 17
    A = torch.rand(N, N) * 5
    A[3, :] = 0.0  # empty row -> inf guard (GCN.py:105)
    out = build_supports(A.unsqueeze(0), "random_walk_diffusion", 2)[0]
    P = _ref_rw_normalize(A)
    ref = _ref_cheb(P.T, 2)
    assert torch.allclose(out, ref, atol=1e-5)


def test_dual_random_walk_shares_identity():
    torch.manual_seed(2)
    N = 11
    A = torch.rand(N, N)
    out = build_supports(A.unsqueeze(0), "dual_random_walk_diffusion", 2)[0]
    assert out.shape[0] == 5
    assert torch.allclose(out[0], torch.eye(N))  # shared order-0 term
    fwd = _ref_cheb(_ref_rw_normalize(A).T, 2)
    bwd = _ref_cheb(_ref_rw_normalize(A.T).T, 2)
    assert torch.allclose(out[:3], fwd, atol=1e-5)
    assert torch.allclose(out[3:], bwd[1:], atol=1e-5)


def test_chebyshev_lambda2_fallback_matches_reference():
    # modern torch has no torch.eig, so the reference ALWAYS rescales with
    # lambda_max = 2 (GCN.py:116-126): L_rescaled = L - I
    torch.manual_seed(3)
    N = 9
    A = torch.rand(N, N) + 0.1
    out = build_supports(A.unsqueeze(0), "chebyshev", 2)[0]
    d = A.sum(1)
    Dm = torch.diag(d.pow(-0.5))
    L = torch.eye(N) - Dm @ A @ Dm
    ref = _ref_cheb(L - torch.eye(N), 2)
    assert torch.allclose(out, ref, atol=1e-5)


def test_localpool():
    torch.manual_seed(4)
    N = 8
    A = torch.rand(N, N) + 0.1
    out = build_supports(A.unsqueeze(0), "localpool", 1)[0]
    d = A.sum(1)
    Dm = torch.diag(d.pow(-0.5))
    ref = torch.eye(N) + Dm @ A @ Dm
    assert out.shape == (1, N, N)
    assert torch.allclose(out[0], ref, atol=1e-5)
