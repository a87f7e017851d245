"""Attribute per-step stock-op GPU time with input shapes (torch.profiler).

Run on a GPU box: python tools/profile_step.py [--nodes N] [--batch B].
Mirrors bench.py's single-rank step; prints the top ops by CUDA time.
"""
import argparse

import torch
from torch.profiler import ProfilerActivity, profile

from mpgcn_amd.graph import build_supports
from mpgcn_amd.models import MPGCN

ap = argparse.ArgumentParser()
ap.add_argument("--nodes", type=int, default=256)
ap.add_argument("--batch", type=int, default=32)
args = ap.parse_args()

dev = "cuda:0"
N, B, H, T = args.nodes, args.batch, 32, 7
S = 3
torch.manual_seed(1234)
model = MPGCN(M=2, K=S, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
              gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N,
              compute_dtype=torch.bfloat16).to(dev)
opt = torch.optim.Adam(model.parameters(), lr=1e-4)
criterion = torch.nn.MSELoss()
T_pool = 64
pool = torch.log1p(20.0 * torch.rand(T_pool, N, N, 1, device=dev))
adj = (torch.rand(N, N, device=dev) < 0.1).float()
G_static = build_supports(adj.unsqueeze(0), "random_walk_diffusion", 2).squeeze(0)
O_dyn_raw = torch.rand(7, N, N, device=dev)
D_dyn_raw = torch.rand(7, N, N, device=dev)


def step(i):
    g = (torch.arange(B, device=dev) * 7 + i) % (T_pool - T - 1)
    x = pool[g.unsqueeze(1) + torch.arange(T, device=dev)]
    y = pool[(g + T).unsqueeze(1) + torch.arange(1, device=dev)]
    key = (g + T) % 7
    G_o = build_supports(O_dyn_raw[key], "random_walk_diffusion", 2)
    G_d = build_supports(D_dyn_raw[key], "random_walk_diffusion", 2)
    loss = criterion(model(x, [G_static, (G_o, G_d)]), y)
    opt.zero_grad(set_to_none=True)
    loss.backward()
    opt.step()


for i in range(3):
    step(i)
torch.cuda.synchronize()
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    for i in range(3):
        step(3 + i)
    torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True).table(
    sort_by="cuda_time_total", row_limit=40, max_src_column_width=60))
