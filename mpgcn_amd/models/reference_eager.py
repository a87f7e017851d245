"""Reference-math eager transcription (numerics oracle + baseline).

Transcribed from the equations in SURVEY.md §2
(reference call sites MPGCN.py:24-50, 54-112). These implement the reference's
direct K^2-pair formulation with stock torch ops — the framework's factored
algorithm and HIP kernels are tested against them.
"""

from __future__ import annotations

import torch
from torch import nn


def bdgcn_pairs_reference(X, Go, Gd, W, b=None, relu=True):
    """Direct K^2-pair 2-D GCN (MPGCN.py:26-49 semantics).

    X: (B,N,N,C); Go/Gd: (S,N,N) or (B,S,N,N); W: (C*S*S, H)."""
    S = Go.shape[-3]
    feats = []
    for o in range(S):
        for d in range(S):
            if Go.dim() == 3:
                m1 = torch.einsum("bncl,nm->bmcl", X, Go[o])
            else:
                m1 = torch.einsum("bncl,bnm->bmcl", X, Go[:, o])
            if Gd.dim() == 3:
                m2 = torch.einsum("bmcl,cd->bmdl", m1, Gd[d])
            else:
                m2 = torch.einsum("bmcl,bcd->bmdl", m1, Gd[:, d])
            feats.append(m2)
    feat = torch.cat(feats, dim=-1)
    out = torch.einsum("bmdk,kh->bmdh", feat, W)
    if b is not None:
        out = out + b
    return torch.relu(out) if relu else out


class MPGCNReference(nn.Module):
    """Full-model oracle: nn.LSTM + pair-formulation BDGCN + Linear/ReLU head,
    module tree named to produce the reference's state_dict keys — used for
    checkpoint-compatibility and numerics-fidelity tests."""

    def __init__(self, M, K, input_dim, hidden, gcn_layers, num_nodes):
        super().__init__()
        self.M, self.K, self.N, self.H = M, K, num_nodes, hidden
        self.gcn_layers = gcn_layers
        self.branch_models = nn.ModuleList()
        for _ in range(M):
            branch = nn.ModuleDict()
            branch["temporal"] = nn.LSTM(input_size=input_dim, hidden_size=hidden,
                                         num_layers=1, batch_first=True)
            spatial = nn.ModuleList()
            for n in range(gcn_layers):
                cur_in = hidden
                lin = nn.Module()
                lin.W = nn.Parameter(torch.empty(cur_in * K * K, hidden))
                nn.init.xavier_normal_(lin.W)
                lin.b = nn.Parameter(torch.zeros(hidden))
                spatial.append(lin)
            branch["spatial"] = spatial
            branch["fc"] = nn.Sequential(nn.Linear(hidden, input_dim), nn.ReLU())
            self.branch_models.append(branch)

    def forward(self, x_seq, G_list):
        B, T, N = x_seq.shape[0], x_seq.shape[1], self.N
        lstm_in = x_seq.permute(0, 2, 3, 1, 4).reshape(B * N * N, T, 1)
        outs = []
        for m in range(self.M):
            branch = self.branch_models[m]
            h0 = torch.zeros(1, B * N * N, self.H, device=x_seq.device,
                             dtype=x_seq.dtype)
            c0 = torch.zeros(1, B * N * N, self.H, device=x_seq.device,
                             dtype=x_seq.dtype)
            lstm_out, _ = branch["temporal"](lstm_in, (h0, c0))
            X = lstm_out[:, -1, :].reshape(B, N, N, self.H)
            G = G_list[m]
            Go, Gd = (G, G) if isinstance(G, torch.Tensor) else G
            for layer in branch["spatial"]:
                X = bdgcn_pairs_reference(X, Go, Gd, layer.W, layer.b, relu=True)
            out = branch["fc"](X)
            outs.append(out)
        return torch.mean(torch.stack(outs, dim=-1), dim=-1).unsqueeze(1)
