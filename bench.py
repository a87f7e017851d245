"""Flagship training-step benchmark (driver contract).

Measures the BASELINE.json headline metric: MPGCN train-step samples/sec on a
256-region OD config — full training semantics per step: on-device dynamic
support construction (the reference rebuilds supports every step,
Model_Trainer.py:106), forward, MSE loss, backward, gradient all-reduce (N>1,
RCCL over xGMI), Adam step. bf16 compute with fp32 master weights; synthetic
OD data (no dataset is bundled with the reference repo) and random-init
weights. Weak scaling: per-GPU batch fixed as N grows.

  python bench.py --gpus N --steps K --warmup W
  (N>1 is launched by the driver via torch.distributed.run, one rank per GPU)
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--nodes", type=int, default=256, help="region count N")
    ap.add_argument("--batch", type=int, default=32, help="per-GPU batch size")
    ap.add_argument("--hidden", type=int, default=32)
    ap.add_argument("--obs-len", type=int, default=7)
    ap.add_argument("--dtype", type=str, default="bf16",
                    choices=["bf16", "float32", "fp8"],
                    help="fp8 = opt-in fp8-forward/bf16-backward mode (the "
                         "contract headline stays bf16)")
    ap.add_argument("--device", type=str, default=None,
                    help="override device (cpu for debug)")
    ap.add_argument("--impl", type=str, default="native", choices=["native", "eager"],
                    help="'eager' runs the reference-math fp32 transcription "
                         "(stock torch ops, K^2-pair formulation) as the baseline")
    ap.add_argument("--branches", type=int, default=2, choices=[2, 3],
                    help="graph perspectives (3 = BASELINE config #2)")
    ap.add_argument("--fusion", type=str, default="mean", choices=["mean", "attention"])
    ap.add_argument("--partition", type=str, default="dp", choices=["dp", "region"],
                    help="multi-rank strategy: data parallel (weak scaling) or "
                         "region partition (activation grid sharded across ranks)")
    ap.add_argument("--graph", action="store_true",
                    help="capture the whole train step in a hipGraph and replay "
                         "it (single-rank; removes host launch gaps — matters "
                         "in the small-batch regime)")
    args = ap.parse_args()

    import torch.distributed as dist

    from mpgcn_amd.graph import build_supports
    from mpgcn_amd.models import MPGCN
    from mpgcn_amd.parallel import GradAllReducer, init_distributed

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))

    if args.device:
        device = args.device
    elif torch.cuda.is_available():
        # ranks share devices round-robin when oversubscribed (2-rank RCCL
        # bring-up on a 1-GPU box)
        local = int(os.environ.get("LOCAL_RANK", 0)) % torch.cuda.device_count()
        device = f"cuda:{local}"
    else:
        device = "cpu"
    ctx = init_distributed(device)
    if device.startswith("cuda"):
        torch.cuda.set_device(device)
    is_cuda = device.startswith("cuda")

    torch.manual_seed(1234 + rank)
    N, B, H, T = args.nodes, args.batch, args.hidden, args.obs_len
    K_order = 2
    kernel = "random_walk_diffusion"
    S = K_order + 1
    fp8 = args.dtype == "fp8" and is_cuda
    cdtype = (torch.bfloat16 if (args.dtype in ("bf16", "fp8") and is_cuda)
              else torch.float32)

    if args.impl == "eager":
        from mpgcn_amd.models.reference_eager import MPGCNReference

        cdtype = torch.float32  # the reference implementation is fp32 eager
        model = MPGCNReference(M=2, K=S, input_dim=1, hidden=H, gcn_layers=3,
                               num_nodes=N).to(device)
    else:
        model = MPGCN(M=args.branches, K=S, input_dim=1, lstm_hidden_dim=H,
                      lstm_num_layers=1, gcn_hidden_dim=H, gcn_num_layers=3,
                      num_nodes=N, compute_dtype=cdtype,
                      fusion=args.fusion, fp8_forward=fp8).to(device)
    use_graph = args.graph and is_cuda and world == 1 and args.impl == "native"
    if args.impl == "native":
        # fused flat-buffer Adam: one kernel per step instead of capturable
        # Adam's ~45 tiny launches (ops/optim.py); must be built before the
        # reducer registers its grad hooks since it repoints p.data/p.grad
        from mpgcn_amd.ops.optim import FlatAdam

        opt = FlatAdam(model.parameters(), lr=1e-4)
    else:
        opt = torch.optim.Adam(model.parameters(), lr=1e-4,
                               capturable=use_graph)
    reducer = GradAllReducer(model, ctx)
    criterion = torch.nn.MSELoss()

    # device-resident synthetic OD pool (log1p-scale magnitudes), windowed by index
    T_pool = 64
    pool = torch.log1p(20.0 * torch.rand(T_pool, N, N, 1, device=device))
    from mpgcn_amd.graph.supports import tag_like

    adj = (torch.rand(N, N, device=device) < 0.1).float()
    _gs = build_supports(adj.unsqueeze(0), kernel, K_order)
    G_static = tag_like(_gs.squeeze(0), _gs)
    _gc = build_supports(torch.rand(1, N, N, device=device), kernel, K_order)
    G_corr = tag_like(_gc.squeeze(0), _gc)  # third perspective (static)
    # raw day-of-week correlation graphs (support build runs per-step, timed)
    O_dyn_raw = torch.rand(7, N, N, device=device)
    D_dyn_raw = torch.rand(7, N, N, device=device)

    region = args.partition == "region" and world > 1
    if region:
        from mpgcn_amd.parallel.region import mpgcn_forward_sharded, shard_dest

    i_buf = torch.zeros((), dtype=torch.long, device=device)

    def step_body(i=None):
        # graph mode reads the device scalar (re-read at every replay);
        # eager mode keeps the host int to avoid extra device round-trips
        g = (torch.arange(B, device=device) * 7 + (i_buf if i is None else i)) \
            % (T_pool - T - 1)
        x = pool[g.unsqueeze(1) + torch.arange(T, device=device)]  # (B,T,N,N,1)
        y = pool[(g + T).unsqueeze(1) + torch.arange(1, device=device)]
        key = (g + T) % 7
        G_o = build_supports(O_dyn_raw[key], kernel, K_order)
        G_d = build_supports(D_dyn_raw[key], kernel, K_order)
        G_list = [G_static, (G_o, G_d)]
        if args.branches == 3 and args.impl == "native":
            G_list.append(G_corr)
        if region:
            xs = shard_dest(x, rank, world)
            y = shard_dest(y, rank, world)
            y_pred = mpgcn_forward_sharded(model, xs, G_list)
        else:
            y_pred = model(x, G_list)
        loss = criterion(y_pred, y)
        # graph mode needs stable grad buffers across replays
        opt.zero_grad(set_to_none=not use_graph)
        loss.backward()
        reducer.finalize()
        opt.step()
        return loss

    if use_graph:
        # warm up on a side stream (allocator + lazy state), then capture one
        # full step — dynamic-support build, forward, backward, Adam — into a
        # hipGraph; replay re-reads i_buf so the data window still advances
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                step_body()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        hip_graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(hip_graph):
            graph_loss = step_body()

    def step(i: int):
        if use_graph:
            i_buf.fill_(i)
            hip_graph.replay()
            return graph_loss
        return step_body(i)

    def barrier_sync():
        if ctx.enabled:
            dist.barrier()
        if is_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        step(i)
    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = step(args.warmup + i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines job time)
    if ctx.enabled:
        t = torch.tensor([elapsed], device=device if is_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    n_gpus = world if world > 1 else (1 if is_cuda else args.gpus)
    global_batch = B * max(world, 1)
    samples_per_sec = global_batch * args.steps / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "train_samples_per_sec",
            "value": round(samples_per_sec, 2),
            "unit": "samples/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("fp8" if fp8 else
                      "bf16" if cdtype == torch.bfloat16 else "float32"),
            "data": "synthetic",
            "config": {
                "model": "MPGCN" if args.impl == "native" else "MPGCN-reference-eager",
                "regions": N,
                "global_batch": global_batch,
                "seq_len": T,
                "hidden": H,
                "kernel": kernel,
                "supports_K": S,
                "gcn_layers": 3,
                "branches": args.branches if args.impl == "native" else 2,
                "parallelism": f"{args.partition}{max(world, 1)}",
                "final_loss": round(float(loss.item()), 5),
            },
        }))
    if ctx.enabled:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
