"""Property-based invariants of the support builders (CPU oracle math).

Hypothesis drives random batch sizes, region counts, orders, sparsity and
magnitudes through build_supports and asserts the structural invariants the
HIP fused builders are separately tested against (tests/test_gpu_kernels.py
compares GPU vs this torch path; these properties pin the torch path itself):

  * T_0 = I for every Chebyshev-family stack (the identity-skip contract);
  * random-walk series: T_1 = P_fwd^T, so every COLUMN of T_1 sums to the
    origin row's mass (1 for non-empty rows, 0 for empty rows) — and empty
    rows never produce inf/nan anywhere in the stack;
  * dual series shares T_0 and its backward block equals the forward series
    of the transposed flow;
  * localpool = I + sym_norm(A) is symmetric for symmetric A;
  * chebyshev terms obey the recurrence T_k = 2 L' T_{k-1} - T_{k-2} exactly;
  * support count always matches get_support_K.
"""

import pytest
import torch

from hypothesis import given, settings, strategies as st

from mpgcn_amd.graph.supports import (
    build_supports,
    get_support_K,
    random_walk_normalize,
)


def _flow(B, N, sparsity, scale, seed, empty_row):
    g = torch.Generator().manual_seed(seed)
    A = torch.rand(B, N, N, generator=g) * scale
    A = A * (torch.rand(B, N, N, generator=g) > sparsity)
    if empty_row:
        A[:, min(1, N - 1), :] = 0.0
    return A


common = dict(
    B=st.integers(1, 4), N=st.integers(2, 12),
    sparsity=st.sampled_from([0.0, 0.5, 0.9]),
    scale=st.sampled_from([1.0, 100.0]),
    seed=st.integers(0, 10_000), empty_row=st.booleans(),
)


@settings(max_examples=40, deadline=None)
@given(order=st.integers(1, 4), **common)
def test_rwd_series_properties(order, B, N, sparsity, scale, seed, empty_row):
    A = _flow(B, N, sparsity, scale, seed, empty_row)
    sup = build_supports(A, "random_walk_diffusion", order)
    assert sup.shape == (B, get_support_K("random_walk_diffusion", order), N, N)
    assert torch.isfinite(sup).all()
    eye = torch.eye(N).expand(B, N, N)
    assert torch.equal(sup[:, 0], eye)
    assert getattr(sup, "_identity_first", False)
    # T_1 = P_fwd^T: column j sums to 1 where row j of A has mass, else 0
    colsum = sup[:, 1].sum(dim=-2)
    mass = (A.sum(dim=-1) > 0).to(colsum.dtype)
    torch.testing.assert_close(colsum, mass, rtol=0, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(order=st.integers(1, 3), **common)
def test_dual_rwd_blocks(order, B, N, sparsity, scale, seed, empty_row):
    A = _flow(B, N, sparsity, scale, seed, empty_row)
    sup = build_supports(A, "dual_random_walk_diffusion", order)
    K = get_support_K("dual_random_walk_diffusion", order)
    assert sup.shape[1] == K == 2 * order + 1
    fwd = build_supports(A, "random_walk_diffusion", order)
    bwd = build_supports(A.transpose(-2, -1), "random_walk_diffusion", order)
    torch.testing.assert_close(sup[:, : order + 1], fwd, rtol=0, atol=0)
    torch.testing.assert_close(sup[:, order + 1:], bwd[:, 1:], rtol=0, atol=0)


@settings(max_examples=25, deadline=None)
@given(order=st.integers(2, 4), **common)
def test_chebyshev_recurrence_exact(order, B, N, sparsity, scale, seed,
                                    empty_row):
    A = _flow(B, N, sparsity, scale, seed, empty_row)
    A = 0.5 * (A + A.transpose(-2, -1))  # Laplacian assumes symmetric flow
    sup = build_supports(A, "chebyshev", order, lambda_max=2.0)
    assert torch.isfinite(sup).all()
    eye = torch.eye(N).expand(B, N, N)
    assert torch.equal(sup[:, 0], eye)
    Lp = sup[:, 1]  # T_1 = rescaled Laplacian itself
    for k in range(2, order + 1):
        want = 2.0 * torch.bmm(Lp, sup[:, k - 1]) - sup[:, k - 2]
        torch.testing.assert_close(sup[:, k], want, rtol=1e-4, atol=1e-4)


@settings(max_examples=25, deadline=None)
@given(**common)
def test_localpool_symmetric(B, N, sparsity, scale, seed, empty_row):
    A = _flow(B, N, sparsity, scale, seed, empty_row)
    A = 0.5 * (A + A.transpose(-2, -1))
    sup = build_supports(A, "localpool", 1)
    assert sup.shape == (B, 1, N, N)
    assert torch.isfinite(sup).all()
    torch.testing.assert_close(sup[:, 0], sup[:, 0].transpose(-2, -1),
                               rtol=1e-5, atol=1e-6)
    # diagonal >= 1 (I plus a nonnegative normalized term)
    diag = sup[:, 0].diagonal(dim1=-2, dim2=-1)
    assert (diag >= 1.0 - 1e-6).all()


@settings(max_examples=30, deadline=None)
@given(**common)
def test_random_walk_normalize_rows(B, N, sparsity, scale, seed, empty_row):
    A = _flow(B, N, sparsity, scale, seed, empty_row)
    P = random_walk_normalize(A)
    assert torch.isfinite(P).all()
    rs = P.sum(dim=-1)
    mass = (A.sum(dim=-1) > 0).to(rs.dtype)
    torch.testing.assert_close(rs, mass, rtol=0, atol=1e-5)


def test_pathological_inputs_stay_finite():
    # all-zero flow, single region, huge magnitudes
    for A in (torch.zeros(2, 5, 5), torch.zeros(1, 1, 1),
              torch.full((1, 4, 4), 1e30)):
        for kt, k in (("random_walk_diffusion", 2), ("localpool", 1),
                      ("dual_random_walk_diffusion", 2)):
            sup = build_supports(A, kt, k)
            assert torch.isfinite(sup).all(), (kt, tuple(A.shape))
