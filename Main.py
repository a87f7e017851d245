"""CLI entrypoint — flag-compatible superset of the reference Main.py:8-37.

All 21 reference flags are accepted with the same names, choices and defaults
(-GPU, -in, -out, -model, -t, -obs, -pred, -norm, -split, -batch, -hidden,
-kernel, -K, -nn, -loss, -optim, -lr, -dr, -epoch, -mode), plus MI355X-native
additions: synthetic data generation (no bundled dataset, no network), compute
dtype, shuffling, and multi-GPU data parallelism via torchrun env vars.

Examples:
  python Main.py -mode train -synthetic-nodes 16 -epoch 3 -GPU cpu
  python Main.py -mode train -synthetic-nodes 256 -dtype bf16 -GPU cuda:0
  torchrun --nproc-per-node 8 Main.py -mode train -synthetic-nodes 256 -dtype bf16
"""

import argparse
import os

import torch


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(description="Run OD Prediction.")
    # reference-compatible flags (Main.py:8-37)
    parser.add_argument("-GPU", "--GPU", type=str, default="cuda:0",
                        help="Device string; 'cpu' for no GPU")
    parser.add_argument("-in", "--input_dir", type=str, default="../data")
    parser.add_argument("-out", "--output_dir", type=str, default="./output")
    parser.add_argument("-model", "--model", type=str, choices=["MPGCN"], default="MPGCN")
    parser.add_argument("-t", "--time_slice", type=int, default=24)
    parser.add_argument("-obs", "--obs_len", type=int, default=7)
    parser.add_argument("-pred", "--pred_len", type=int, default=7)
    parser.add_argument("-norm", "--norm", type=str,
                        choices=["none", "minmax", "std"], default="none")
    parser.add_argument("-split", "--split_ratio", type=float, nargs="+",
                        default=[6.4, 1.6, 2])
    parser.add_argument("-batch", "--batch_size", type=int, default=4)
    parser.add_argument("-hidden", "--hidden_dim", type=int, default=32)
    parser.add_argument("-kernel", "--kernel_type", type=str,
                        choices=["chebyshev", "localpool", "random_walk_diffusion",
                                 "dual_random_walk_diffusion"],
                        default="random_walk_diffusion")
    parser.add_argument("-K", "--cheby_order", type=int, default=2)
    parser.add_argument("-nn", "--nn_layers", type=int, default=2)
    parser.add_argument("-loss", "--loss", type=str,
                        choices=["MSE", "MAE", "Huber"], default="MSE")
    parser.add_argument("-optim", "--optimizer", type=str, default="Adam")
    parser.add_argument("-lr", "--learn_rate", type=float, default=1e-4)
    parser.add_argument("-dr", "--decay_rate", type=float, default=0)
    parser.add_argument("-epoch", "--num_epochs", type=int, default=200)
    parser.add_argument("-mode", "--mode", type=str, choices=["train", "test"],
                        default="train")
    # MI355X-native additions
    parser.add_argument("-synthetic-nodes", "--synthetic_nodes", type=int, default=0,
                        help="Generate a synthetic N-region OD dataset instead of "
                             "loading npz files from --input_dir")
    parser.add_argument("-synthetic-days", "--synthetic_days", type=int, default=425)
    parser.add_argument("-dtype", "--compute_dtype", type=str,
                        choices=["float32", "bf16", "fp8"], default="float32",
                        help="Compute dtype on GPU (fp32 master weights either "
                             "way); fp8 = fp8-forward/bf16-backward mode")
    parser.add_argument("-shuffle", "--shuffle", action="store_true",
                        help="Shuffle training batches (reference default: off)")
    parser.add_argument("-seed", "--seed", type=int, default=0)
    parser.add_argument("-partition", "--partition", type=str,
                        choices=["dp", "region"], default="dp",
                        help="multi-GPU strategy: data parallel (default) or "
                             "region partition (shard the N x N activation grid)")
    parser.add_argument("-M", "--perspectives", type=int, choices=[2, 3], default=2,
                        help="graph perspectives: 2 (reference: adjacency + "
                             "dynamic OD-correlation) or 3 (+ static OD-correlation)")
    parser.add_argument("-fusion", "--fusion", type=str,
                        choices=["mean", "attention"], default="mean",
                        help="branch fusion: arithmetic mean (reference) or "
                             "learned attention weights")
    parser.add_argument("-ref-quirks", "--ref_quirks", action="store_true",
                        help="reproduce the reference's dynamic D-graph "
                             "computation exactly (Data_Container_OD.py:53-56 "
                             "column/row mixing) for bit-parity validation")
    parser.add_argument("-resume", "--resume", action="store_true",
                        help="resume training from the extended checkpoint "
                             "({model}_od.resume.pkl) if present")
    return parser


def main():
    from mpgcn_amd.data import DataGenerator, DataInput
    from mpgcn_amd.parallel import init_distributed, rank_watchdog
    from mpgcn_amd.train import ModelTrainer

    params = build_parser().parse_args().__dict__
    os.makedirs(params["output_dir"], exist_ok=True)

    if params["mode"] == "train":
        params["pred_len"] = 1  # train single-step model (Main.py:44-45)

    torch.manual_seed(params["seed"])

    device = params["GPU"]
    if device.startswith("cuda") and not torch.cuda.is_available():
        print("[mpgcn] no GPU visible, falling back to cpu")
        device = "cpu"
    ctx = init_distributed(device)
    if ctx.enabled and device.startswith("cuda") and torch.cuda.is_available():
        device = f"cuda:{ctx.local_rank % torch.cuda.device_count()}"
    params["device"] = device

    data_input = DataInput(params=params)
    data = data_input.load_data()
    params["N"] = data["OD"].shape[1]

    data_generator = DataGenerator(
        obs_len=params["obs_len"], pred_len=params["pred_len"],
        data_split_ratio=params["split_ratio"],
    )
    # region partition shards the activation grid, not the sample axis:
    # every rank sees the full batch stream
    dl_rank = ctx.rank if params["partition"] == "dp" else 0
    dl_world = ctx.world_size if params["partition"] == "dp" else 1
    data_loader = data_generator.get_data_loader(
        data=data, params=params, device=device,
        rank=dl_rank, world_size=dl_world,
    )

    trainer = ModelTrainer(params=params, data=data,
                           data_container=data_input, dist_ctx=ctx)

    with rank_watchdog(ctx):
        if params["mode"] == "train":
            trainer.train(data_loader=data_loader, modes=["train", "validate"])
        else:
            trainer.test(data_loader=data_loader, modes=["train", "test"])


if __name__ == "__main__":
    main()
