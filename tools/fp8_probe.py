"""fp8 e4m3 mode-2 probe: numerics + timing vs the bf16 production kernel.

Measurement-only (docs/ROADMAP.md byte-reduction lever): quantizes the
flagship mode-2 operands to OCP fp8, runs the same axis engine at BK=128
(half the staged bytes per K element), and reports relative error vs an
fp32 einsum plus kernel time vs the bf16 path. Run on a GPU box.
"""
import torch

from mpgcn_amd import ops

dev = "cuda:0"
ext = ops.get_ext()
torch.manual_seed(0)

B, N, S, H = 32, 256, 3, 32
V32 = torch.randn(B, N, N * S, H, device=dev) / (N * S) ** 0.25
A32 = torch.randn(N, N * S, device=dev) / (N * S) ** 0.25
ref = torch.einsum("dk,bmkh->bmdh", A32, V32)

V16, A16 = V32.bfloat16(), A32.bfloat16()
V8 = V32.to(torch.float8_e4m3fn)
A8 = A32.to(torch.float8_e4m3fn)

y16 = ext.bdgcn_mode2(V16, A16, None, False, N, S)
y8 = ext.bdgcn_mode2_fp8(V8, A8, None, False, N, S)


def relerr(y):
    return ((y.float() - ref).norm() / ref.norm()).item()


print(f"rel err bf16 {relerr(y16):.4f}  fp8 {relerr(y8):.4f}")


def time_it(fn, iters=50):
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000


t16 = time_it(lambda: ext.bdgcn_mode2(V16, A16, None, False, N, S))
t8 = time_it(lambda: ext.bdgcn_mode2_fp8(V8, A8, None, False, N, S))
fl = 2 * B * N * (N * S) * (N * H)
print(f"bf16 {t16:8.1f} us  ({fl/t16/1e6:6.1f} TF/s)")
print(f"fp8  {t8:8.1f} us  ({fl/t8/1e6:6.1f} TF/s)  speedup {t16/t8:.2f}x")

# ---- mode-1 shape (K = N = 256: short-K regime) ----
C = 32
X32 = torch.randn(B, N, N, C, device=dev) / N**0.25
G32 = torch.randn(3, N, N, device=dev) / N**0.25
ref1 = torch.einsum("onm,bndl->bmdol", G32, X32)
GT32 = G32.transpose(-2, -1).contiguous()
u16 = ext.bdgcn_mode1(X32.bfloat16(), GT32.bfloat16())
u8 = ext.bdgcn_mode1_fp8(X32.to(torch.float8_e4m3fn),
                         GT32.to(torch.float8_e4m3fn))
print(f"mode1 rel err bf16 {((u16.float()-ref1).norm()/ref1.norm()).item():.4f}"
      f"  fp8 {((u8.float()-ref1).norm()/ref1.norm()).item():.4f}")
X16, GT16 = X32.bfloat16(), GT32.bfloat16()
X8m, GT8m = X32.to(torch.float8_e4m3fn), GT32.to(torch.float8_e4m3fn)
t16 = time_it(lambda: ext.bdgcn_mode1(X16, GT16))
t8 = time_it(lambda: ext.bdgcn_mode1_fp8(X8m, GT8m))
fl1 = 2 * B * 3 * N * N * (N * C)
print(f"mode1 bf16 {t16:8.1f} us ({fl1/t16/1e6:6.1f} TF/s)")
print(f"mode1 fp8  {t8:8.1f} us ({fl1/t8/1e6:6.1f} TF/s)  speedup {t16/t8:.2f}x")
