"""MPGCN model: state_dict key compatibility, numerics fidelity vs the
nn.LSTM-based reference transcription, checkpoint round trip."""

import torch

from mpgcn_amd.graph import build_supports
from mpgcn_amd.models import MPGCN
from tests.oracle import MPGCNReference


def _expected_keys(M=2, layers=3):
    keys = []
    for m in range(M):
        keys += [f"branch_models.{m}.temporal.{p}" for p in
                 ("weight_ih_l0", "weight_hh_l0", "bias_ih_l0", "bias_hh_l0")]
        for n in range(layers):
            keys += [f"branch_models.{m}.spatial.{n}.W",
                     f"branch_models.{m}.spatial.{n}.b"]
        keys += [f"branch_models.{m}.fc.0.weight", f"branch_models.{m}.fc.0.bias"]
    return set(keys)


def _make(N=10, K=3, H=16):
    return MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                 gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N)


def test_state_dict_keys_match_reference():
    model = _make()
    assert set(model.state_dict().keys()) == _expected_keys()


def test_state_dict_shapes_match_reference_oracle():
    model = _make(N=10, K=3, H=16)
    oracle = MPGCNReference(M=2, K=3, input_dim=1, hidden=16, gcn_layers=3,
                            num_nodes=10)
    ours = model.state_dict()
    theirs = oracle.state_dict()
    assert set(ours.keys()) == set(theirs.keys())
    for k in ours:
        assert ours[k].shape == theirs[k].shape, k


def test_forward_matches_reference_oracle_after_weight_copy():
    torch.manual_seed(0)
    N, K, H, B, T = 8, 3, 16, 2, 5
    model = _make(N=N, K=K, H=H)
    oracle = MPGCNReference(M=2, K=K, input_dim=1, hidden=H, gcn_layers=3,
                            num_nodes=N)
    oracle.load_state_dict(model.state_dict())  # checkpoint compatibility

    x = torch.rand(B, T, N, N, 1)
    flow = torch.rand(B, N, N)
    Gs = build_supports(torch.rand(1, N, N), "random_walk_diffusion", K - 1)[0]
    Go = build_supports(flow, "random_walk_diffusion", K - 1)
    Gd = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", K - 1)

    with torch.no_grad():
        ours = model(x, [Gs, (Go, Gd)])
        ref = oracle(x, [Gs, (Go, Gd)])
    assert ours.shape == (B, 1, N, N, 1)
    assert torch.allclose(ours, ref, atol=1e-4), (ours - ref).abs().max()


def test_checkpoint_roundtrip_from_oracle():
    """A checkpoint written by the reference-style model loads into ours."""
    torch.manual_seed(1)
    oracle = MPGCNReference(M=2, K=2, input_dim=1, hidden=16, gcn_layers=3,
                            num_nodes=6)
    ckpt = {"epoch": 7, "state_dict": oracle.state_dict()}
    model = _make(N=6, K=2, H=16)
    model.load_state_dict(ckpt["state_dict"])  # must not raise
    for k, v in model.state_dict().items():
        assert torch.equal(v, ckpt["state_dict"][k])


def test_backward_produces_grads():
    torch.manual_seed(2)
    N, K, H = 6, 2, 16
    model = _make(N=N, K=K, H=H)
    x = torch.rand(1, 4, N, N, 1)
    flow = torch.rand(1, N, N)
    Gs = build_supports(torch.rand(1, N, N), "random_walk_diffusion", K - 1)[0]
    Go = build_supports(flow, "random_walk_diffusion", K - 1)
    out = model(x, [Gs, (Go, Go)])
    out.sum().backward()
    for name, p in model.named_parameters():
        assert p.grad is not None, name
        assert torch.isfinite(p.grad).all(), name


def test_three_perspectives_attention_fusion():
    """M=3 + learned attention fusion (BASELINE config #2 / north-star
    'multi-graph attention fusion'); the default M=2 mean fusion keeps the
    reference state_dict exactly (tested above)."""
    torch.manual_seed(3)
    N, K, H = 8, 3, 16
    model = MPGCN(M=3, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                  gcn_hidden_dim=H, gcn_num_layers=2, num_nodes=N,
                  fusion="attention")
    assert "fusion_w" in model.state_dict()
    x = torch.rand(2, 5, N, N, 1)
    flow = torch.rand(2, N, N)
    Gs = build_supports(torch.rand(1, N, N), "random_walk_diffusion", K - 1)[0]
    Gc = build_supports(torch.rand(1, N, N), "random_walk_diffusion", K - 1)[0]
    Go = build_supports(flow, "random_walk_diffusion", K - 1)
    out = model(x, [Gs, (Go, Go), Gc])
    assert out.shape == (2, 1, N, N, 1)
    out.sum().backward()
    assert model.fusion_w.grad is not None

    # mean fusion (default) must not add parameters
    m2 = MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
               gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N)
    assert "fusion_w" not in m2.state_dict()
