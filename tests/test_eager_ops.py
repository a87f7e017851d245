"""Factored BDGCN algorithm and eager LSTM vs reference math."""

import torch

from mpgcn_amd.ops import eager
from tests.oracle import bdgcn_pairs_reference


def test_factored_bdgcn_static_matches_pairs():
    torch.manual_seed(0)
    B, N, C, H, S = 3, 13, 8, 8, 3
    X = torch.randn(B, N, N, C)
    G = torch.randn(S, N, N)
    W = torch.randn(C * S * S, H)
    b = torch.randn(H)
    ref = bdgcn_pairs_reference(X, G, G, W, b, relu=True)
    out = eager.bdgcn_layer_eager(X, G, G, W, b, "relu")
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


def test_factored_bdgcn_dynamic_matches_pairs():
    torch.manual_seed(1)
    B, N, C, H, S = 2, 9, 4, 6, 2
    X = torch.randn(B, N, N, C)
    Go = torch.randn(B, S, N, N)
    Gd = torch.randn(B, S, N, N)
    W = torch.randn(C * S * S, H)
    ref = bdgcn_pairs_reference(X, Go, Gd, W, None, relu=False)
    out = eager.bdgcn_layer_eager(X, Go, Gd, W, None, "none")
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


def test_factored_bdgcn_gradients_match_pairs():
    torch.manual_seed(2)
    B, N, C, H, S = 2, 7, 4, 4, 2
    X = torch.randn(B, N, N, C, requires_grad=True)
    G = torch.randn(S, N, N)
    W = torch.randn(C * S * S, H, requires_grad=True)
    b = torch.randn(H, requires_grad=True)

    out = eager.bdgcn_layer_eager(X, G, G, W, b, "relu")
    out.square().sum().backward()
    gX, gW, gb = X.grad.clone(), W.grad.clone(), b.grad.clone()
    X.grad = W.grad = b.grad = None

    ref = bdgcn_pairs_reference(X, G, G, W, b, relu=True)
    ref.square().sum().backward()
    assert torch.allclose(gX, X.grad, atol=1e-3)
    assert torch.allclose(gW, W.grad, atol=1e-3)
    assert torch.allclose(gb, b.grad, atol=1e-3)


def test_eager_lstm_matches_nn_lstm():
    torch.manual_seed(3)
    R, T, H = 50, 7, 32
    x = torch.randn(R, T, 1)
    lstm = torch.nn.LSTM(1, H, num_layers=1, batch_first=True)
    ref_out, _ = lstm(x, (torch.zeros(1, R, H), torch.zeros(1, R, H)))
    out, h_t, c_t = eager.lstm_forward_eager(
        x, lstm.weight_ih_l0, lstm.weight_hh_l0, lstm.bias_ih_l0, lstm.bias_hh_l0
    )
    assert torch.allclose(out, ref_out, atol=1e-5)
    assert torch.allclose(h_t, ref_out[:, -1, :], atol=1e-5)


def test_projection_weight_reorder_roundtrip():
    S, C, H = 3, 4, 5
    W = torch.arange(S * S * C * H, dtype=torch.float32).reshape(S * S * C, H)
    Wre = eager.reorder_projection_weight(W, S, C)
    # Wre[o*C + l, s*H + h] == W[(o*S + s)*C + l, h]
    for o in range(S):
        for s in range(S):
            for l in range(C):
                for h in range(H):
                    assert Wre[o * C + l, s * H + h] == W[(o * S + s) * C + l, h]


def test_gcn1d_matches_reference_math():
    """1-D GCN library op vs the GCN.py:23-43 equations."""
    from mpgcn_amd.models import GCN

    torch.manual_seed(5)
    K, N, C, H, B = 3, 11, 4, 6, 2
    m = GCN(K=K, input_dim=C, hidden_dim=H)
    G = torch.randn(K, N, N)
    x = torch.randn(B, N, C)
    out = m(G, x)
    sup = torch.cat([torch.einsum("ij,bjp->bip", G[k], x) for k in range(K)], dim=-1)
    ref = torch.relu(sup @ m.W + m.b)
    assert torch.allclose(out, ref, atol=1e-5)
