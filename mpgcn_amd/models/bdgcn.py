"""BDGCN: 2-D graph convolution on an N x N OD matrix (the core op).

Capability- and parameter-compatible with the reference BDGCN (MPGCN.py:6-50):
parameter names (`W`, `b`), shapes (W: (input_dim * K^2, hidden_dim)), init
(xavier-normal W, constant-0 b) and math are identical; the execution uses the
factored mode-1 / projection / mode-2 algorithm (mpgcn_amd/ops/eager.py) on
hand-written gfx950 HIP kernels when on GPU.
"""

from __future__ import annotations

import torch
from torch import nn

from mpgcn_amd.ops import GraphOperator, bdgcn_layer, bdgcn_layer_fp8
from mpgcn_amd.ops.functional import make_fp8_state


class BDGCN(nn.Module):
    def __init__(self, K: int, input_dim: int, hidden_dim: int,
                 use_bias: bool = True, activation: str = "relu"):
        super().__init__()
        self.K = K
        self.input_dim = input_dim
        self.hidden_dim = hidden_dim
        self.use_bias = use_bias
        if activation not in ("relu", "none", None):
            raise ValueError("BDGCN supports activation 'relu' or 'none'")
        self.relu = activation == "relu"
        self.W = nn.Parameter(torch.empty(input_dim * K * K, hidden_dim))
        nn.init.xavier_normal_(self.W)
        if use_bias:
            self.b = nn.Parameter(torch.zeros(hidden_dim))
        else:
            self.register_parameter("b", None)

    def forward(self, X: torch.Tensor, gop: GraphOperator, fp8: bool = False,
                X8: torch.Tensor | None = None, emit_twin: bool = True):
        """X: (B, N, N, input_dim) -> (B, N, N, hidden_dim).

        fp8=True runs the fp8-forward/bf16-backward path and returns
        (Y, Y8_twin) so the caller can thread the fp8 twin into the next
        layer without a quantize pass (ops/functional.py); emit_twin=False
        (last layer) returns Y alone and skips the twin write."""
        W = self.W.to(X.dtype if X.dtype != torch.float8_e4m3fn else torch.bfloat16)
        if fp8:
            st = getattr(self, "_fp8_state", None)
            if st is None or st["amax_u"].device != X.device:
                st = self._fp8_state = make_fp8_state(X.device)
            return bdgcn_layer_fp8(X, W, self.b, gop, relu=self.relu, X8=X8,
                                   emit_twin=emit_twin, fp8_state=st)
        return bdgcn_layer(X, W, self.b, gop, relu=self.relu)

    def extra_repr(self) -> str:
        return f"K={self.K}, in={self.input_dim}, hidden={self.hidden_dim}, bias={self.use_bias}"
