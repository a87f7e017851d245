"""Diagnostic: fp8 delayed-scaling adaptation under small (MSE-mean-scale)
gradients — prints per-iteration gradient norms and the per-layer scale
state next to the bf16 reference."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from mpgcn_amd.graph import build_supports
from mpgcn_amd.ops import GraphOperator, bdgcn_layer, bdgcn_layer_fp8
from mpgcn_amd.ops.functional import make_fp8_state

dev = "cuda:0"
N, S, C, H, B = 256, 3, 32, 32, 2
torch.manual_seed(0)
X = torch.rand(B, N, N, C, device=dev).bfloat16()
Go = build_supports(torch.rand(1, N, N, device=dev),
                    "random_walk_diffusion", 2).squeeze(0).bfloat16().contiguous()
import sys as _s
if "--full" in _s.argv:
    gop = GraphOperator(Go, Go, id_first=False)
else:
    Go._identity_first = True
    gop = GraphOperator(Go, Go)
W = (0.1 * torch.randn(C * S * S, H, device=dev)).bfloat16()
st = make_fp8_state(dev)

Xb = X.clone().requires_grad_(True)
Wb = W.clone().requires_grad_(True)
Yb = bdgcn_layer(Xb, Wb, None, gop, relu=True)
(Yb.float() ** 2).mean().backward()
print("bf16 dX norm %.3e dW norm %.3e" % (Xb.grad.float().norm(), Wb.grad.float().norm()))

for it in range(3):
    Xi = X.clone().requires_grad_(True)
    Wi = W.clone().requires_grad_(True)
    Y, _ = bdgcn_layer_fp8(Xi, Wi, None, gop, relu=True, fp8_state=st)
    (Y.float() ** 2).mean().backward()
    torch.cuda.synchronize()
    print("it%d dX %.3e dW %.3e | amax_y %.3e scale_y %.3e inv_y %.3e | "
          "amax_u %.3e scale_u %.3e inv_u %.3e" % (
              it, Xi.grad.float().norm(), Wi.grad.float().norm(),
              st["amax_y"].item(), st["scale_y"].item(), st["inv_y"].item(),
              st["amax_u"].item(), st["scale_u"].item(), st["inv_u"].item()))
