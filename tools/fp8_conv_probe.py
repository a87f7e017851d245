"""Minimal trainer-shaped convergence probe: full MPGCN (M=2, dynamic
graphs), MSE + Adam over a small cycling batch set — loss curves for bf16,
fp8, and fp8-forward-only (MPGCN_FP8_BWD=0 bisect)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from mpgcn_amd.graph import build_supports
from mpgcn_amd.graph.supports import tag_like
from mpgcn_amd.models import MPGCN


def run(tag, fp8, steps=60):
    dev = "cuda:0"
    torch.manual_seed(0)
    N, B, H, T, S = 256, 32, 32, 7, 3
    pool = torch.log1p(20.0 * torch.rand(16, N, N, 1, device=dev))
    _gs = build_supports((torch.rand(1, N, N, device=dev) < 0.1).float(),
                         "random_walk_diffusion", 2)
    Gs = tag_like(_gs.squeeze(0), _gs)
    O_raw = torch.rand(7, N, N, device=dev)
    D_raw = torch.rand(7, N, N, device=dev)
    torch.manual_seed(1)
    model = MPGCN(M=2, K=S, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                  gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N,
                  compute_dtype=torch.bfloat16, fp8_forward=fp8).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    losses = []
    for i in range(steps):
        g = (torch.arange(B, device=dev) * 3 + i) % (16 - T - 1)
        x = pool[g.unsqueeze(1) + torch.arange(T, device=dev)]
        y = pool[(g + T).unsqueeze(1) + torch.arange(1, device=dev)]
        key = (g + T) % 7
        Go = build_supports(O_raw[key], "random_walk_diffusion", 2)
        Gd = build_supports(D_raw[key], "random_walk_diffusion", 2)
        out = model(x, [Gs, (Go, Gd)])
        loss = torch.nn.functional.mse_loss(out, y)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        if fp8 and i in (0, 1, 2, 5, 20) and os.environ.get("MPGCN_FP8_DUMP"):
            torch.cuda.synchronize()
            for bi, br in enumerate(model.branch_models):
                for li, lay in enumerate(br["spatial"]):
                    st = getattr(lay, "_fp8_state", None)
                    g = lay.W.grad
                    print(f"  step{i} b{bi}L{li} dW {0 if g is None else g.float().norm():.3e} "
                          f"amax_y {st['amax_y'].item():.3e} scale_y {st['scale_y'].item():.3e} "
                          f"amax_u {st['amax_u'].item():.3e} scale_u {st['scale_u'].item():.3e}"
                          if st else f"  step{i} b{bi}L{li} NO STATE")
        opt.step()
        losses.append(loss.item())
    print(tag, " ".join(f"{v:.4f}" for v in losses[::6]), "final", f"{losses[-1]:.4f}")


if __name__ == "__main__":
    run("bf16    ", False)
    run("fp8     ", True)
    os.environ["MPGCN_FP8_BWD"] = "0"
    run("fp8-fwd ", True)
