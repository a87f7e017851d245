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
        B, N, C = x.shape
        if (x.is_cuda and x.dtype in (torch.bfloat16, torch.float32)
                and self.hidden_dim <= 128 and self.K * C <= 2048):
            # HIP path (K5): the K-support contraction is a mode-1 axis GEMM
            # with a singleton destination axis, and the projection + bias +
            # act is one fused row_gemm — the same MFMA kernels as BDGCN
            # (ext.hip bdgcn_mode1 / row_gemm; eager einsum is the oracle).
            from mpgcn_amd.ops.functional import _ops

            ext = _ops.get_ext()
            # the kernel's A operand is consumed row-major as A[m, n] with n
            # contracted; the 1-D GCN wants sum_j G[i, j] x[j], so G itself
            # (NOT transposed — BDGCN's mode-1 passes G^T because its einsum
            # contracts the FIRST graph index, GCN.py:34 vs MPGCN.py:30)
            A = G.to(x.dtype).contiguous()
            U = ext.bdgcn_mode1(x.reshape(B, N, 1, C).contiguous(), A, False)
            # U: (B, N, 1, K, C) — concat over supports is the natural layout
            bias_f = self.b.float().contiguous() if self.b is not None else None
            out = ext.row_gemm(
                U.reshape(B * N, self.K * C),
                self.W.to(x.dtype).contiguous(), bias_f, self.relu,
            )
            return out.view(B, N, self.hidden_dim)
        # eager fallback (CPU / exotic shapes): reference math (GCN.py:34-44)
        sup = torch.einsum("kij,bjp->bkip", G, x)
        cat = sup.permute(0, 2, 1, 3).reshape(B, N, self.K * C)
        out = cat @ self.W
        if self.b is not None:
            out = out + self.b
        return torch.relu(out) if self.relu else out

    def __repr__(self):
        return f"GCN({self.K} * input {self.input_dim} -> hidden {self.hidden_dim})"
