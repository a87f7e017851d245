"ovhd spy: attribute large copy_/clone/contiguous/to/cat calls in one flagship step to mpgcn_amd source lines (TorchFunctionMode; run on GPU via gpurun)."
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, traceback
from torch.overrides import TorchFunctionMode
from mpgcn_amd.models import MPGCN
from mpgcn_amd.graph.supports import build_supports, tag_like
from mpgcn_amd.ops.optim import FlatAdam
from collections import Counter

dev = "cuda:0"
N, H, B, T = 256, 32, 32, 7
torch.manual_seed(0)
model = MPGCN(M=2, K=3, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
              gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N,
              compute_dtype=torch.bfloat16).to(dev)
opt = FlatAdam(model.parameters(), lr=1e-4)
crit = torch.nn.MSELoss()
pool = torch.log1p(20.0 * torch.rand(64, N, N, 1, device=dev))
adj = (torch.rand(N, N, device=dev) < 0.1).float()
_gs = build_supports(adj.unsqueeze(0), "random_walk_diffusion", 2)
G_static = tag_like(_gs.squeeze(0), _gs)
O_dyn = torch.rand(7, N, N, device=dev)
D_dyn = torch.rand(7, N, N, device=dev)

stats = Counter(); nbytes = Counter()

class Spy(TorchFunctionMode):
    def __torch_function__(self, func, types, args=(), kwargs=None):
        kwargs = kwargs or {}
        name = getattr(func, "__name__", str(func))
        if name in ("copy_", "clone", "contiguous", "to", "_to_copy", "cat"):
            t = args[0]
            if isinstance(t, torch.Tensor) and t.numel() * t.element_size() > 256_000:
                site = "?"
                for fr in traceback.extract_stack()[::-1]:
                    if "mpgcn_amd" in fr.filename:
                        site = f"{fr.filename.split('mpgcn_amd/')[-1]}:{fr.lineno}"
                        break
                key = (name, site)
                stats[key] += 1
                nbytes[key] += t.numel() * t.element_size()
        return func(*args, **kwargs)

def step(i):
    g = (torch.arange(B, device=dev) * 7 + i) % 56
    x = pool[g.unsqueeze(1) + torch.arange(T, device=dev)]
    y = pool[(g + T).unsqueeze(1) + torch.arange(1, device=dev)]
    key = (g + T) % 7
    G_o = build_supports(O_dyn[key], "random_walk_diffusion", 2)
    G_d = build_supports(D_dyn[key], "random_walk_diffusion", 2)
    y_pred = model(x, [G_static, (G_o, G_d)])
    loss = crit(y_pred, y)
    opt.zero_grad(); loss.backward(); opt.step()

step(0); torch.cuda.synchronize()
with Spy():
    step(1)
torch.cuda.synchronize()
for (name, site), c in sorted(stats.items(), key=lambda kv: -nbytes[kv[0]])[:22]:
    print(f"{nbytes[(name,site)]/1e6:8.2f} MB  x{c:3d}  {name:12s} {site}")
