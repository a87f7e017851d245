"""Region-partition engine: 2-rank gloo sharded forward/backward must match
the unsharded computation (SURVEY.md §7 "region-partition correctness")."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from mpgcn_amd.graph import build_supports
from mpgcn_amd.models import MPGCN

N, K, H, B, T = 8, 3, 16, 2, 5
P = 2


def _inputs():
    torch.manual_seed(0)
    x = torch.rand(B, T, N, N, 1)
    y = torch.rand(B, 1, N, N, 1)
    flow = torch.rand(B, N, N)
    Gs = build_supports(torch.rand(1, N, N), "random_walk_diffusion", K - 1)[0]
    Go = build_supports(flow, "random_walk_diffusion", K - 1)
    Gd = build_supports(flow.transpose(-2, -1), "random_walk_diffusion", K - 1)
    return x, y, Gs, Go, Gd


def _model():
    torch.manual_seed(1)
    return MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H, lstm_num_layers=1,
                 gcn_hidden_dim=H, gcn_num_layers=3, num_nodes=N)


def _worker(rank, file_name, out_file):
    from mpgcn_amd.parallel.region import mpgcn_forward_sharded, shard_dest

    os.environ.update(RANK=str(rank), WORLD_SIZE=str(P), LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", init_method=f"file://{file_name}",
                            rank=rank, world_size=P)
    model = _model()
    x, y, Gs, Go, Gd = _inputs()
    xs = shard_dest(x, rank, P)
    ys = shard_dest(y, rank, P)

    out = mpgcn_forward_sharded(model, xs, [Gs, (Go, Gd)])
    loss = torch.nn.functional.mse_loss(out, ys)
    loss.backward()
    # average weight grads across ranks (weights replicated)
    for p_ in model.parameters():
        dist.all_reduce(p_.grad)
        p_.grad /= P

    if rank == 0:
        torch.save(
            {"out": out.detach(),
             "grads": {n: q.grad.clone() for n, q in model.named_parameters()}},
            out_file,
        )
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_sharded_forward_backward_matches_unsharded(tmp_path):
    file_name = str(tmp_path / "pg")
    out_file = str(tmp_path / "rank0.pt")
    ctxm = mp.get_context("spawn")
    procs = [ctxm.Process(target=_worker, args=(r, file_name, out_file))
             for r in range(P)]
    for p_ in procs:
        p_.start()
    for p_ in procs:
        p_.join(timeout=240)
        assert p_.exitcode == 0
    got = torch.load(out_file, weights_only=True)

    model = _model()
    x, y, Gs, Go, Gd = _inputs()
    ref = model(x, [Gs, (Go, Gd)])
    torch.nn.functional.mse_loss(ref, y).backward()

    # rank 0 output shard == full output's first dest slice
    Nl = N // P
    torch.testing.assert_close(got["out"], ref[..., :Nl, :].detach(),
                               atol=1e-5, rtol=1e-5)
    for n, p_ in model.named_parameters():
        torch.testing.assert_close(got["grads"][n], p_.grad, atol=1e-5,
                                   rtol=1e-4, msg=n)


def test_a2a_roundtrip_single_rank(tmp_path):
    """Shard algebra sanity without a process group: P=1 all-to-all is identity
    up to the sharding permutation."""
    import torch.distributed as dist

    from mpgcn_amd.parallel.region import dest_to_origin, origin_to_dest

    store_file = str(tmp_path / "pg1")
    dist.init_process_group("gloo", init_method=f"file://{store_file}",
                            rank=0, world_size=1)
    try:
        x = torch.arange(2 * 4 * 4 * 3, dtype=torch.float32).reshape(2, 4, 4, 3)
        o = dest_to_origin(x)
        assert torch.equal(o, x)  # P=1: permutation is identity
        back = origin_to_dest(o)
        assert torch.equal(back, x)
    finally:
        dist.destroy_process_group()


def test_split_halves_match_full_layer_rectangular():
    """The layer split at the all-to-all seam (mode1_proj | mode2_bias_act)
    reproduces the fused layer on rectangular shards, forward and backward —
    single-process, with the re-shard emulated by slice/cat."""
    from mpgcn_amd.ops import GraphOperator, eager, mode1_proj, mode2_bias_act

    torch.manual_seed(4)
    S, C, Hd, Bb, Pp = 3, 4, 16, 2, 2
    Nl = N // Pp
    X = torch.randn(Bb, N, N, C, requires_grad=True)
    Go = torch.randn(S, N, N)
    Gd = torch.randn(S, N, N)
    W = torch.randn(C * S * S, Hd, requires_grad=True)
    bias = torch.randn(Hd, requires_grad=True)
    gop = GraphOperator(Go, Gd)

    ref = eager.bdgcn_layer_eager(X, Go, Gd, W, bias, "relu")
    ref.square().sum().backward()
    gX, gW, gb = X.grad.clone(), W.grad.clone(), bias.grad.clone()
    X.grad = W.grad = bias.grad = None

    Vs = [mode1_proj(X[:, :, p * Nl:(p + 1) * Nl, :], W, gop) for p in range(Pp)]
    Vfull = torch.cat(Vs, dim=2)  # (B, N, N, S*H) — emulated all-to-all
    Ys = []
    for p in range(Pp):
        Vo = Vfull[:, p * Nl:(p + 1) * Nl].reshape(Bb, Nl, N, S, Hd)
        Ys.append(mode2_bias_act(Vo, bias, gop, True))
    out = torch.cat(Ys, dim=1)
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-5)
    out.square().sum().backward()
    torch.testing.assert_close(X.grad, gX, atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(W.grad, gW, atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(bias.grad, gb, atol=1e-5, rtol=1e-4)


def test_sharded_attention_fusion_matches_unsharded(tmp_path):
    """-fusion attention + -partition region: the sharded forward applies the
    learned softmax fusion (it is pointwise, so shard-compatible) instead of
    silently degrading to mean; P=1 sharded output must equal the full model."""
    from mpgcn_amd.parallel.region import mpgcn_forward_sharded

    store_file = str(tmp_path / "pg_att")
    dist.init_process_group("gloo", init_method=f"file://{store_file}",
                            rank=0, world_size=1)
    try:
        torch.manual_seed(2)
        model = MPGCN(M=2, K=K, input_dim=1, lstm_hidden_dim=H,
                      lstm_num_layers=1, gcn_hidden_dim=H, gcn_num_layers=2,
                      num_nodes=N, fusion="attention")
        with torch.no_grad():
            model.fusion_w.copy_(torch.tensor([0.7, -0.3]))  # non-uniform
        x, y, Gs, Go, Gd = _inputs()
        out_sharded = mpgcn_forward_sharded(model, x, [Gs, (Go, Gd)])
        ref = model(x, [Gs, (Go, Gd)])
        torch.testing.assert_close(out_sharded, ref, atol=1e-6, rtol=1e-5)
        # fusion_w must receive gradient through the sharded path
        out_sharded.square().sum().backward()
        assert model.fusion_w.grad is not None
        assert model.fusion_w.grad.abs().sum() > 0
    finally:
        dist.destroy_process_group()


def _trainer_worker(rank, file_name, out_dir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(P), LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", init_method=f"file://{file_name}",
                            rank=rank, world_size=P)
    from mpgcn_amd.data import DataGenerator, DataInput
    from mpgcn_amd.parallel.ddp import DistContext
    from mpgcn_amd.train import ModelTrainer

    params = {
        "model": "MPGCN", "synthetic_nodes": N, "synthetic_days": 60,
        "seed": 0, "split_ratio": [7, 1.5, 1.5], "norm": "none",
        "obs_len": 5, "pred_len": 1, "batch_size": 2, "hidden_dim": 16,
        "kernel_type": "random_walk_diffusion", "cheby_order": 2,
        "loss": "MSE", "optimizer": "Adam", "learn_rate": 1e-3,
        "decay_rate": 0, "num_epochs": 1, "output_dir": out_dir,
        "device": "cpu", "partition": "region", "N": N,
    }
    di = DataInput(params=params)
    data = di.load_data()
    gen = DataGenerator(obs_len=5, pred_len=1, data_split_ratio=[7, 1.5, 1.5])
    # region partition: all ranks iterate the full batch stream
    loaders = gen.get_data_loader(data=data, params=params, device="cpu",
                                  rank=0, world_size=1)
    ctx = DistContext(rank=rank, world_size=P, local_rank=rank, backend="gloo")
    trainer = ModelTrainer(params=params, data=data, data_container=di,
                           dist_ctx=ctx)
    trainer.train(data_loader=loaders, modes=["train", "validate"])
    # region-sharded evaluation: each rank rolls out its destination shard,
    # metric statistics all-reduce, rank 0 writes the scores line
    trainer.test(data_loader=loaders, modes=["test"])
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_region_trainer_end_to_end(tmp_path):
    """ModelTrainer with -partition region on a 2-rank gloo group: epochs run,
    weight grads stay synchronized (replicated model), rank 0 checkpoints."""
    file_name = str(tmp_path / "pg_tr")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir, exist_ok=True)
    ctxm = mp.get_context("spawn")
    procs = [ctxm.Process(target=_trainer_worker, args=(r, file_name, out_dir))
             for r in range(P)]
    for p_ in procs:
        p_.start()
    for p_ in procs:
        p_.join(timeout=240)
        assert p_.exitcode == 0
    assert os.path.exists(os.path.join(out_dir, "MPGCN_od.pkl"))
    scores = open(os.path.join(out_dir, "MPGCN_prediction_scores.txt")).read()
    line = scores.splitlines()[0]
    assert line.startswith("test, MSE, RMSE, MAE, MAPE, ")
    assert all(v == v for v in map(float, line.split(", ")[5:]))  # finite


def test_region_world_size_divisibility_error():
    import torch.distributed as dist

    from mpgcn_amd.parallel.region import mpgcn_forward_sharded

    store = "/tmp/_pgdiv"
    import os as _os
    if _os.path.exists(store):
        _os.remove(store)
    dist.init_process_group("gloo", init_method=f"file://{store}",
                            rank=0, world_size=1)
    try:
        model = _model()
        x = torch.rand(1, T, 7, 7, 1)  # 7 regions, any P>... N=7 with P=1 ok;
        # use a fake group world via monkeypatched N % P: P=1 divides, so
        # instead check the error path directly
        import pytest as _pytest

        from mpgcn_amd.parallel import region as reg

        class _FakeDist:
            @staticmethod
            def get_world_size(group=None):
                return 2

        orig = reg.dist
        reg.dist = _FakeDist
        try:
            with _pytest.raises(ValueError, match="divisible"):
                mpgcn_forward_sharded(model, x, None)
        finally:
            reg.dist = orig
    finally:
        dist.destroy_process_group()
