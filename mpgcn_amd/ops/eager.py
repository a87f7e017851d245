"""Eager (pure-PyTorch) implementations of the core ops.

These serve two roles:
  1. CPU execution path (this framework runs end-to-end without a GPU, e.g. the
     16-region plumbing config of BASELINE.json).
  2. The reference math for unit tests of the HIP kernels.

The BDGCN layer here uses the *factored* algorithm (see ``bdgcn_layer_eager``):
algebraically identical to the reference's K^2-pair formulation (MPGCN.py:24-50)
but with K mode-1 + K mode-2 axis products instead of K^2 of each — a ~3x FLOP
reduction at K = 3. The HIP path implements the same factorization.

Derivation. The reference computes, for every support pair (o, s):
    Z_{o,s}[b,m,d,l] = sum_c Gd[s,c,d] * (sum_n Go[o,n,m] * X[b,n,c,l])
then concatenates over (o, s) on the channel axis and projects:
    H[b,m,d,h] = sum_{o,s,l} Z_{o,s}[b,m,d,l] * W[(o*K+s)*C + l, h]        (+ bias, act)
Since the destination-axis product and the projection are both linear, swap them:
    U_o[b,m,c,l]   = sum_n Go[o,n,m] * X[b,n,c,l]                  (K mode-1 products)
    V_s[b,m,c,h]   = sum_{o,l} U_o[b,m,c,l] * W[(o*K+s)*C + l, h]  (one flat GEMM)
    H[b,m,d,h]     = sum_{s,c} Gd[s,c,d] * V_s[b,m,c,h]            (K mode-2 products)
which is exactly H above, term for term.
"""

from __future__ import annotations

import torch


def reorder_projection_weight(W: torch.Tensor, S: int, C: int) -> torch.Tensor:
    """(C*S*S, H) reference-layout W -> (S*C, S*H) factored-layout Wre.

    Reference row index k = (o*S + s)*C + l (concat order of MPGCN.py:28-44);
    Wre[o*C + l, s*H + h] = W[(o*S + s)*C + l, h].
    """
    H = W.shape[1]
    return W.view(S, S, C, H).permute(0, 2, 1, 3).reshape(S * C, S * H)


def mode1_apply(X: torch.Tensor, G: torch.Tensor) -> torch.Tensor:
    """Origin-axis graph product. X:(B,N,N,C), G:(S,N,N) or (B,S,N,N).

    U[b,m,d,o,l] = sum_n G[(b,)o,n,m] X[b,n,d,l]; output layout (B, N, N, S, C)
    so that (o, l) is the contiguous trailing pair the projection GEMM consumes.
    """
    if G.dim() == 3:
        return torch.einsum("onm,bndl->bmdol", G, X)
    return torch.einsum("bonm,bndl->bmdol", G, X)


def mode2_apply(V: torch.Tensor, G: torch.Tensor) -> torch.Tensor:
    """Destination-axis graph product summed over supports.

    V:(B,N,N,S,H), G:(S,N,N) or (B,S,N,N);
    Y[b,m,d,h] = sum_{s,c} G[(b,)s,c,d] V[b,m,c,s,h].
    """
    if G.dim() == 3:
        return torch.einsum("scd,bmcsh->bmdh", G, V)
    return torch.einsum("bscd,bmcsh->bmdh", G, V)


def bdgcn_layer_eager(
    X: torch.Tensor,
    Go: torch.Tensor,
    Gd: torch.Tensor,
    W: torch.Tensor,
    bias: torch.Tensor | None,
    activation: str = "relu",
) -> torch.Tensor:
    """Full 2-D GCN layer (factored algorithm). See module docstring.

    X: (B, N, N, C); Go/Gd: (S, N, N) static or (B, S, N, N) dynamic;
    W: (C*S*S, H) in the reference's concat layout; bias: (H,) or None.
    """
    S = Go.shape[-3]
    C = X.shape[-1]
    B, N = X.shape[0], X.shape[1]
    Hdim = W.shape[1]

    U = mode1_apply(X, Go)  # (B, N, N, S, C)
    Wre = reorder_projection_weight(W, S, C)
    V = (U.reshape(B * N * N, S * C) @ Wre).view(B, N, N, S, Hdim)
    Y = mode2_apply(V, Gd)
    if bias is not None:
        Y = Y + bias
    if activation == "relu":
        Y = torch.relu(Y)
    elif activation not in (None, "none", "linear"):
        raise ValueError(f"unsupported activation {activation!r}")
    return Y


def lstm_forward_eager(
    x: torch.Tensor,
    w_ih: torch.Tensor,
    w_hh: torch.Tensor,
    b_ih: torch.Tensor,
    b_hh: torch.Tensor,
    h0: torch.Tensor | None = None,
    c0: torch.Tensor | None = None,
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Single-layer batch-first LSTM, torch gate order (i, f, g, o).

    x: (R, T, I); w_ih: (4H, I); w_hh: (4H, H). Returns (out (R,T,H), h_T, c_T).
    Matches nn.LSTM(num_layers=1, batch_first=True) numerics.
    """
    R, T, _ = x.shape
    Hd = w_hh.shape[1]
    h = x.new_zeros(R, Hd) if h0 is None else h0
    c = x.new_zeros(R, Hd) if c0 is None else c0
    bias = b_ih + b_hh
    outs = []
    for t in range(T):
        gates = x[:, t, :] @ w_ih.t() + h @ w_hh.t() + bias
        i, f, g, o = gates.chunk(4, dim=1)
        i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
        g = torch.tanh(g)
        c = f * c + i * g
        h = o * torch.tanh(c)
        outs.append(h)
    return torch.stack(outs, dim=1), h, c
