"""Property-based tests (hypothesis): the factored BDGCN algorithm and the
support builders hold over randomized shapes/values, not just the fixtures."""

import torch
from hypothesis import given, settings, strategies as st

from mpgcn_amd.graph import build_supports, get_support_K
from mpgcn_amd.ops import eager
from mpgcn_amd.models.reference_eager import bdgcn_pairs_reference


@settings(max_examples=25, deadline=None)
@given(
    n=st.integers(3, 24),
    c=st.integers(1, 8),
    h=st.integers(1, 8),
    s=st.integers(1, 4),
    b=st.integers(1, 3),
    dyn=st.booleans(),
    relu=st.booleans(),
    seed=st.integers(0, 2**16),
)
def test_factored_equals_pairs(n, c, h, s, b, dyn, relu, seed):
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(b, n, n, c, generator=g)
    shape = (b, s, n, n) if dyn else (s, n, n)
    Go = torch.randn(*shape, generator=g)
    Gd = torch.randn(*shape, generator=g)
    W = torch.randn(c * s * s, h, generator=g)
    bias = torch.randn(h, generator=g)
    ref = bdgcn_pairs_reference(X, Go, Gd, W, bias, relu=relu)
    out = eager.bdgcn_layer_eager(X, Go, Gd, W, bias, "relu" if relu else "none")
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4), (out - ref).abs().max()


@settings(max_examples=20, deadline=None)
@given(
    kernel=st.sampled_from(["localpool", "chebyshev", "random_walk_diffusion",
                            "dual_random_walk_diffusion"]),
    order=st.integers(1, 4),
    n=st.integers(2, 16),
    b=st.integers(1, 4),
    seed=st.integers(0, 2**16),
)
def test_support_count_and_finiteness(kernel, order, n, b, seed):
    if kernel == "localpool":
        order = 1
    g = torch.Generator().manual_seed(seed)
    # non-negative flow with some guaranteed empty rows
    flow = torch.rand(b, n, n, generator=g)
    flow[:, 0, :] = 0.0
    out = build_supports(flow, kernel, order)
    assert out.shape == (b, get_support_K(kernel, order), n, n)
    assert torch.isfinite(out).all()


@settings(max_examples=20, deadline=None)
@given(
    t=st.integers(2, 6),
    r=st.integers(1, 40),
    h=st.integers(4, 24),
    seed=st.integers(0, 2**16),
)
def test_eager_lstm_matches_nn(t, r, h, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(r, t, 1, generator=g)
    lstm = torch.nn.LSTM(1, h, num_layers=1, batch_first=True)
    ref, _ = lstm(x)
    out, _, _ = eager.lstm_forward_eager(
        x, lstm.weight_ih_l0, lstm.weight_hh_l0, lstm.bias_ih_l0, lstm.bias_hh_l0
    )
    assert torch.allclose(out, ref, atol=1e-5)


def test_training_is_seed_deterministic(tmp_path):
    """Same seed -> bit-identical checkpoint (the logical race detector of
    SURVEY.md §5: any nondeterministic reduction/order bug breaks this)."""
    from tests.test_trainer import _setup

    states = []
    for _ in range(2):
        params, trainer, loaders = _setup(tmp_path, num_epochs=2)
        trainer.train(loaders, ["train", "validate"])
        states.append({k: v.clone() for k, v in trainer.model.state_dict().items()})
    for k in states[0]:
        assert torch.equal(states[0][k], states[1][k]), k


@settings(max_examples=15, deadline=None)
@given(
    n=st.sampled_from([4, 6, 8, 12]),
    p=st.sampled_from([1, 2, 4]),
    c=st.integers(1, 6),
    h=st.integers(2, 8),
    s=st.integers(1, 3),
    seed=st.integers(0, 2**16),
)
def test_region_halves_property(n, p, c, h, s, seed):
    """mode1_proj | manual re-shard | mode2_bias_act over P rectangular shards
    equals the fused eager layer for random shapes (the region-partition
    algebra, mpgcn_amd/parallel/region.py)."""
    if n % p != 0:
        return
    from mpgcn_amd.ops import GraphOperator, mode1_proj, mode2_bias_act

    g = torch.Generator().manual_seed(seed)
    nl = n // p
    X = torch.randn(2, n, n, c, generator=g)
    Go = torch.randn(s, n, n, generator=g)
    Gd = torch.randn(s, n, n, generator=g)
    W = torch.randn(c * s * s, h, generator=g)
    bias = torch.randn(h, generator=g)
    ref = eager.bdgcn_layer_eager(X, Go, Gd, W, bias, "relu")
    gop = GraphOperator(Go, Gd)
    V = torch.cat([mode1_proj(X[:, :, q * nl:(q + 1) * nl, :], W, gop)
                   for q in range(p)], dim=2)
    out = torch.cat([
        mode2_bias_act(V[:, q * nl:(q + 1) * nl].reshape(2, nl, n, s, h),
                       bias, gop, True)
        for q in range(p)], dim=1)
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
