"""Per-kernel microbenchmarks at the flagship bench shapes (N=256, B=32, C=H=32,
S=3, T=7). Prints one line per kernel: time, achieved bandwidth, achieved TFLOP/s.
Run on a GPU box:  python bench_kernels.py [--reps 20]
"""

from __future__ import annotations

import argparse
import json

import torch


def timeit(fn, reps=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e3  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=20)
    ap.add_argument("--nodes", type=int, default=256)
    ap.add_argument("--batch", type=int, default=32)
    args = ap.parse_args()

    from mpgcn_amd import ops

    ext = ops.get_ext()
    dev = "cuda:0"
    N, B, C, H, S, T = args.nodes, args.batch, 32, 32, 3, 7
    R = B * N * N
    dt = torch.bfloat16
    torch.manual_seed(0)

    X = torch.randn(B, N, N, C, device=dev, dtype=dt)
    GT = torch.randn(S, N, N, device=dev, dtype=dt)
    U = torch.randn(B, N, N, S, C, device=dev, dtype=dt)
    V = torch.randn(B, N, N * S, H, device=dev, dtype=dt)
    A2T = torch.randn(N, N * S, device=dev, dtype=dt)
    A2 = torch.randn(N * S, N, device=dev, dtype=dt)
    A3T = torch.randn(N, S * N, device=dev, dtype=dt)
    dY = torch.randn(B, N, N, H, device=dev, dtype=dt)
    Wre = torch.randn(S * C, S * H, device=dev, dtype=dt)
    bias = torch.randn(H, device=dev)

    GB = 1e9
    el = 2  # bf16 bytes

    results = {}

    def rec(name, us, bytes_moved, flops):
        results[name] = {
            "us": round(us, 1),
            "TB/s": round(bytes_moved / (us * 1e-6) / 1e12, 2),
            "TF/s": round(flops / (us * 1e-6) / 1e12, 1),
        }
        print(f"{name:24s} {us:9.1f} us   {results[name]['TB/s']:6.2f} TB/s   "
              f"{results[name]['TF/s']:7.1f} TF/s")

    # mode1: U = G^T X per (b,o)
    us = timeit(lambda: ext.bdgcn_mode1(X, GT), args.reps)
    rec("mode1 (fwd)", us, el * (B * S * N * N * C + B * N * N * C + B * N * N * S * C),
        2.0 * B * S * N * N * N * C)

    us = timeit(lambda: ext.bdgcn_mode2(V, A2T, bias, True, N, S), args.reps)
    rec("mode2 (fwd)", us, el * (B * N * N * S * H + B * N * N * H),
        2.0 * B * N * N * (N * S) * H)

    us = timeit(lambda: ext.bdgcn_mode2_bwd(dY, A2, S), args.reps)
    rec("mode2_bwd (dV)", us, el * (B * N * N * H + B * N * N * S * H),
        2.0 * B * N * S * N * N * H)

    us = timeit(lambda: ext.bdgcn_mode1_bwd(U, A3T), args.reps)
    rec("mode1_bwd (dX)", us, el * (B * N * N * S * C + B * N * N * C),
        2.0 * B * N * S * N * N * C)

    Uflat = U.reshape(R, S * C)
    us = timeit(lambda: ext.row_gemm(Uflat, Wre, None, False), args.reps)
    rec("row_gemm 96x96", us, el * (R * S * C + R * S * H), 2.0 * R * S * C * S * H)

    dG = torch.randn(T * R, 4 * H, device=dev, dtype=dt)
    hprev = torch.randn(T * R, H, device=dev, dtype=dt)
    xT = torch.randn(T * R, 1, device=dev, dtype=dt)
    us = timeit(lambda: ext.red_gemm(dG, hprev, True, xT, 1, 0), args.reps)
    rec("red_gemm 128x32 (lstm)", us, el * (T * R * 4 * H + T * R * H),
        2.0 * T * R * 4 * H * H)

    dVflat = torch.randn(R, S * H, device=dev, dtype=dt)
    us = timeit(lambda: ext.red_gemm(dVflat, Uflat, False, None, 0, 0), args.reps)
    rec("red_gemm 96x96 (dWre)", us, el * (R * S * H + R * S * C),
        2.0 * R * S * H * S * C)

    # fused LSTM step
    xseq = torch.randn(R, T, device=dev, dtype=dt)
    h0 = torch.zeros(R, H, device=dev, dtype=dt)
    c0 = torch.zeros(R, H, device=dev, dtype=torch.float32)
    h1 = torch.empty_like(h0)
    c1 = torch.empty_like(c0)
    g1 = torch.empty(R, 4 * H, device=dev, dtype=dt)
    whh = torch.randn(4 * H, H, device=dev, dtype=dt)
    wih = torch.randn(4 * H, device=dev)
    bb = torch.randn(4 * H, device=dev)
    us = timeit(lambda: ext.lstm_step_fwd(xseq, T, 0, h0, c0, whh, wih, bb, h1, c1, g1),
                args.reps)
    rec("lstm_step_fwd", us,
        el * (R * H + R + R * H + R * 4 * H) + 4 * (R * H + R * H),
        2.0 * R * H * 4 * H)

    dh = torch.randn(R, H, device=dev, dtype=dt)
    dcin = torch.randn(R, H, device=dev, dtype=torch.float32)
    dg_out = torch.empty(R, 4 * H, device=dev, dtype=dt)
    dc_out = torch.empty(R, H, device=dev, dtype=torch.float32)
    us = timeit(lambda: ext.lstm_step_bwd(dh, dcin, g1, c0, c1, dg_out, dc_out),
                args.reps)
    rec("lstm_step_bwd", us,
        el * (R * H + R * 4 * H + R * 4 * H) + 4 * (3 * R * H + R * H), 0)

    print(json.dumps(results))


if __name__ == "__main__":
    main()
