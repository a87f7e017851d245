"""Run one kernel in a loop for PMC profiling. Usage: bench_one.py [mode2|mode1|row|red]"""
import sys
import torch
from mpgcn_amd import ops

ext = ops.get_ext()
dev = "cuda:0"
N, B, C, H, S = 256, 32, 32, 32, 3
dt = torch.bfloat16
torch.manual_seed(0)
which = sys.argv[1] if len(sys.argv) > 1 else "mode2"
V = torch.randn(B, N, N * S, H, device=dev, dtype=dt)
A2T = torch.randn(N, N * S, device=dev, dtype=dt)
bias = torch.randn(H, device=dev)
X = torch.randn(B, N, N, C, device=dev, dtype=dt)
GT = torch.randn(S, N, N, device=dev, dtype=dt)
R = B * N * N
Uflat = torch.randn(R, S * C, device=dev, dtype=dt)
Wre = torch.randn(S * C, S * H, device=dev, dtype=dt)
xlf = torch.randn(R, 8, device=dev, dtype=dt)
whh = torch.randn(4 * H, H, device=dev, dtype=dt) * 0.2
whh2 = whh.t().contiguous()
wih = torch.randn(4 * H, device=dev)
bb = torch.randn(4 * H, device=dev)
dh = torch.randn(R, H, device=dev, dtype=dt)
for _ in range(10):
    if which == "mode2":
        ext.bdgcn_mode2(V, A2T, bias, True, N, S)
    elif which == "mode1":
        ext.bdgcn_mode1(X, GT)
    elif which == "row":
        ext.row_gemm(Uflat, Wre, None, False)
    elif which == "lstmf":
        ext.lstm_fused_fwd(xlf, 7, whh, wih, bb)
    elif which == "lstmb":
        ext.lstm_fused_bwd(xlf, 7, whh, whh2, wih, bb, dh, False)
torch.cuda.synchronize()
print("done", which)
