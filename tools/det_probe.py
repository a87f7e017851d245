"""Bitwise-reproducibility probe: call each HIP op twice on identical inputs
and report any drift; then two full model steps. Run on a GPU box."""
import torch

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mpgcn_amd import ops
from mpgcn_amd.graph import build_supports
from mpgcn_amd.models import MPGCN

dev = "cuda:0"
ext = ops.get_ext()
torch.use_deterministic_algorithms(True)


def cmp(tag, f):
    torch.manual_seed(0)
    a = f()
    torch.manual_seed(0)
    b = f()
    if isinstance(a, torch.Tensor):
        a, b = [a], [b]
    bad = [i for i, (x, y) in enumerate(zip(a, b))
           if x is not None and not torch.equal(x, y)]
    print(f"{tag}: {'DRIFT at ' + str(bad) if bad else 'ok'}")


R = 200_000
x = torch.randn(R, 8, device=dev, dtype=torch.bfloat16)
whh = torch.randn(128, 32, device=dev, dtype=torch.bfloat16) * 0.1
wih = torch.randn(128, device=dev) * 0.1
bias = torch.randn(128, device=dev) * 0.1
dh = torch.randn(R, 32, device=dev, dtype=torch.bfloat16)
cmp("lstm_fused_fwd", lambda: ext.lstm_fused_fwd(x, 0, 7, whh, wih, bias, None, None, False))
dxbuf = torch.empty_like(x)
cmp("lstm_fused_bwd", lambda: ext.lstm_fused_bwd(
    x, 0, 7, whh, whh.t().contiguous(), wih, bias, dh, None, None, None,
    False, dxbuf))

dH = torch.randn(65536, 32, device=dev, dtype=torch.bfloat16)
Y = torch.randn(65536, 32, device=dev, dtype=torch.bfloat16)
cmp("relu_bwd_colsum(det)", lambda: ext.relu_bwd_colsum(dH, Y, True))

X2 = torch.randn(100_000, 96, device=dev, dtype=torch.bfloat16)
Y2 = torch.randn(100_000, 96, device=dev, dtype=torch.bfloat16)
xv = torch.randn(100_000, 1, device=dev, dtype=torch.bfloat16)
cmp("red_gemm(det)", lambda: ext.red_gemm(X2, Y2, True, xv, 1, 0))

N = 48
Xa = torch.randn(4, N, N, 32, device=dev, dtype=torch.bfloat16)
G = torch.randn(3, N, N, device=dev, dtype=torch.bfloat16)
cmp("mode1", lambda: ext.bdgcn_mode1(Xa, G.transpose(-2, -1).contiguous()))

# full model: 2 trials of 4 steps, bitwise loss compare
def trial():
    torch.manual_seed(21)
    torch.cuda.manual_seed_all(21)
    model = MPGCN(M=2, K=3, input_dim=1, lstm_hidden_dim=32, lstm_num_layers=1,
                  gcn_hidden_dim=32, gcn_num_layers=3, num_nodes=48,
                  compute_dtype=torch.bfloat16).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    xs = torch.rand(4, 7, 48, 48, 1, device=dev)
    ys = torch.rand(4, 1, 48, 48, 1, device=dev)
    flow = torch.rand(4, 48, 48, device=dev)
    Gs = build_supports(torch.rand(1, 48, 48, device=dev), "random_walk_diffusion", 2)[0]
    Go = build_supports(flow, "random_walk_diffusion", 2)
    Gd = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", 2)
    losses, gr = [], None
    for _ in range(4):
        loss = torch.nn.functional.mse_loss(model(xs, [Gs, (Go, Gd)]), ys)
        opt.zero_grad()
        loss.backward()
        if gr is None:
            gr = {n: p.grad.clone() for n, p in model.named_parameters()}
        opt.step()
        losses.append(loss.item())
    return losses, gr

l1, g1 = trial()
l2, g2 = trial()
print("model losses:", "ok" if l1 == l2 else f"DRIFT {l1} vs {l2}")
for n in g1:
    if not torch.equal(g1[n], g2[n]):
        print("  grad drift (step1):", n, (g1[n] - g2[n]).abs().max().item())
