import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from mpgcn_amd.graph import build_supports
from mpgcn_amd.ops import GraphOperator
dev = "cuda:0"
B, N = 32, 256
raw_o = torch.rand(B, N, N, device=dev)
raw_d = torch.rand(B, N, N, device=dev)

def step(fp8=False):
    Go = build_supports(raw_o, "random_walk_diffusion", 2)
    Gd = build_supports(raw_d, "random_walk_diffusion", 2)
    Go_c = Go.to(torch.bfloat16); Go_c._identity_first = True
    Gd_c = Gd.to(torch.bfloat16); Gd_c._identity_first = True
    gop = GraphOperator(Go_c, Gd_c)
    _ = gop.GoT, gop.A2T, gop.A2, gop.A3T
    if fp8:
        _ = gop.GoT8, gop.A2T8, gop.A28, gop.A3T8
    return gop

def timeit(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3

# build alone vs build+layouts
def build_only():
    build_supports(raw_o, "random_walk_diffusion", 2)
    build_supports(raw_d, "random_walk_diffusion", 2)
print("build only        : %.3f ms" % timeit(build_only))
print("build+cast+layouts: %.3f ms" % timeit(lambda: step(False)))
print("  + fp8 twins     : %.3f ms" % timeit(lambda: step(True)))
