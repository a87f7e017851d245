"""1-D multi-support graph convolution (library op).

Parity with the reference's GCN class (GCN.py:6-45) — dead code on the MPGCN
path there (never instantiated), but part of the library
surface (lineage with ST-MGCN), so provided as a working module: parameter
names (W, b), shapes (W: (K*input_dim, hidden)), init and math match.

  forward(G (K, N, N), x (B, N, input_dim)) -> (B, N, hidden)
  out = act(concat_k(G_k @ x) @ W + b)
"""

from __future__ import annotations

import torch
from torch import nn


class GCN(nn.Module):
    def __init__(self, K: int, input_dim: int, hidden_dim: int, bias: bool = True,
                 activation: str = "relu"):
        super().__init__()
        self.K = K
        self.input_dim = input_dim
        self.hidden_dim = hidden_dim
        self.use_bias = bias
        self.relu = activation == "relu"
        self.W = nn.Parameter(torch.empty(K * input_dim, hidden_dim))
        nn.init.xavier_normal_(self.W)
        if bias:
            self.b = nn.Parameter(torch.zeros(hidden_dim))
        else:
            self.register_parameter("b", None)

    def forward(self, G: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
        assert self.K == G.shape[0]
        # support products: (K, N, N) x (B, N, C) -> (B, N, K*C)
        sup = torch.einsum("kij,bjp->bkip", G, x)
        B, K, N, C = sup.shape
        cat = sup.permute(0, 2, 1, 3).reshape(B, N, K * C)
        out = cat @ self.W
        if self.b is not None:
            out = out + self.b
        return torch.relu(out) if self.relu else out

    def __repr__(self):
        return f"GCN({self.K} * input {self.input_dim} -> hidden {self.hidden_dim})"
