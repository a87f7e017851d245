"""Standalone lstm_fused fwd/bwd timing at the flagship shape (R = 2M, T=7)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch

from mpgcn_amd import ops

ext = ops.get_ext()
dev = "cuda:0"
R, T = 32 * 256 * 256, 7
x = torch.randn(R, 8, device=dev, dtype=torch.bfloat16)
whh = (torch.randn(128, 32, device=dev) * 0.1).bfloat16()
whhT = whh.t().contiguous()
wih = torch.randn(128, device=dev) * 0.1
bias = torch.randn(128, device=dev) * 0.1
dh = torch.randn(R, 32, device=dev, dtype=torch.bfloat16)

def timeit(fn, n=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3

fwd = timeit(lambda: ext.lstm_fused_fwd(x, 0, T, whh, wih, bias, None, None, False))
bwd = timeit(lambda: ext.lstm_fused_bwd(x, 0, T, whh, whhT, wih, bias, dh,
                                        None, None, None, False, None))
print(f"lstm_fused_fwd {fwd:.3f} ms   lstm_fused_bwd {bwd:.3f} ms  (R={R}, T={T})")
