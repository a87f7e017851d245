"""Data layer: windowing indices, split arithmetic, day-of-week keys — vs the
reference's copy-based semantics (Data_Container_OD.py:83-163)."""

import torch

from mpgcn_amd.data import DataGenerator, DataInput


def _params(N=10, T=60, batch=4, obs=7, pred=1):
    return {
        "synthetic_nodes": N, "synthetic_days": T, "norm": "none",
        "split_ratio": [6.4, 1.6, 2], "batch_size": batch,
        "obs_len": obs, "pred_len": pred, "seed": 0,
    }


def _reference_windows(OD, obs, pred):
    """get_feats (Data_Container_OD.py:158-163): materialized copies."""
    xs, ys = [], []
    for i in range(obs, OD.shape[0] - pred):
        xs.append(OD[i - obs:i])
        ys.append(OD[i:i + pred])
    return torch.stack(xs), torch.stack(ys)


def test_windowing_matches_reference_copies():
    p = _params()
    data = DataInput(p).load_data()
    gen = DataGenerator(p["obs_len"], p["pred_len"], p["split_ratio"])
    loaders = gen.get_data_loader(data, p)

    x_ref, y_ref = _reference_windows(data["OD"], p["obs_len"], p["pred_len"])
    mode_len = gen.split2len(x_ref.shape[0])

    # concatenate all batches per mode and compare against reference slices
    starts = {"train": 0, "validate": mode_len["train"],
              "test": mode_len["train"] + mode_len["validate"]}
    for mode in ("train", "validate", "test"):
        xs, ys, keys = [], [], []
        for x, y, O_g, D_g in loaders[mode]:
            xs.append(x)
            ys.append(y)
            assert O_g.shape == (x.shape[0], 10, 10)
            assert D_g.shape == (x.shape[0], 10, 10)
            keys.append(O_g)
        xs = torch.cat(xs)
        ys = torch.cat(ys)
        s, L = starts[mode], mode_len[mode]
        assert xs.shape[0] == L
        assert torch.equal(xs, x_ref[s:s + L])
        assert torch.equal(ys, y_ref[s:s + L])


def test_day_of_week_key_matches_reference():
    """timestamp_query (Data_Container_OD.py:97-108): key = (obs + global) % 7."""
    p = _params(batch=3)
    data = DataInput(p).load_data()
    gen = DataGenerator(p["obs_len"], p["pred_len"], p["split_ratio"])
    loaders = gen.get_data_loader(data, p)
    O_dyn = data["O_dyn_G"]

    n_samples = data["OD"].shape[0] - p["obs_len"] - p["pred_len"]
    mode_len = gen.split2len(n_samples)
    g = mode_len["train"]  # first validate sample's global index
    batch = next(iter(loaders["validate"]))
    for i in range(batch[2].shape[0]):
        key = (p["obs_len"] + g + i) % 7
        assert torch.equal(batch[2][i], O_dyn[:, :, key])


def test_split_lengths_match_reference():
    gen = DataGenerator(7, 1, [6.4, 1.6, 2])
    ml = gen.split2len(100)
    assert ml["validate"] == int(1.6 / 10 * 100)
    assert ml["test"] == int(2 / 10 * 100)
    assert ml["train"] == 100 - ml["validate"] - ml["test"]


def test_normalization_stats_and_inverse():
    p = _params()
    p["norm"] = "minmax"
    di = DataInput(p)
    data = di.load_data()
    OD = data["OD"]
    assert OD.min() >= 0 and OD.max() <= 1.0001
    rec = di.minmax_denormalize(OD)
    assert rec.max() > 1.0  # inverse transform restores the log1p scale

    p2 = _params()
    p2["norm"] = "std"
    di2 = DataInput(p2)
    OD2 = di2.load_data()["OD"]
    assert abs(OD2.mean().item()) < 1e-3
    assert abs(OD2.std().item() - 1.0) < 1e-3


def test_rank_sharding_partitions_train_set():
    p = _params(batch=2)
    data = DataInput(p).load_data()
    gen = DataGenerator(p["obs_len"], p["pred_len"], p["split_ratio"])
    l0 = gen.get_data_loader(data, p, rank=0, world_size=2)["train"]
    l1 = gen.get_data_loader(data, p, rank=1, world_size=2)["train"]
    x0 = torch.cat([b[0] for b in l0])
    x1 = torch.cat([b[0] for b in l1])
    full = gen.get_data_loader(data, p)["train"]
    xf = torch.cat([b[0] for b in full])
    per = xf.shape[0] // 2
    assert torch.equal(x0, xf[:per])
    assert torch.equal(x1, xf[per:2 * per])


# ---- property-based sweep of the windowing/split/key invariants ----------
from hypothesis import given, settings, strategies as st  # noqa: E402


@settings(max_examples=25, deadline=None)
@given(obs=st.integers(2, 10), pred=st.integers(1, 4),
       T=st.integers(40, 90), N=st.integers(3, 10),
       batch=st.integers(1, 8), seed=st.integers(0, 1000))
def test_windowing_property_sweep(obs, pred, T, N, batch, seed):
    """For ANY obs/pred/T/N/batch: every mode's concatenated batches equal the
    reference's materialized window copies at the right split offsets, and
    mode lengths partition the sample count with the reference's arithmetic
    (Data_Container_OD.py:83-163)."""
    p = _params(N=N, T=T, batch=batch, obs=obs, pred=pred)
    p["seed"] = seed
    data = DataInput(p).load_data()
    gen = DataGenerator(obs, pred, p["split_ratio"])
    loaders = gen.get_data_loader(data, p)
    x_ref, y_ref = _reference_windows(data["OD"], obs, pred)
    n = x_ref.shape[0]
    ml = gen.split2len(n)
    assert ml["train"] + ml["validate"] + ml["test"] == n
    assert ml["train"] == n - int(1.6 / 10 * n) - int(2 / 10 * n)
    start = 0
    for mode in ("train", "validate", "test"):
        xs = [x for x, *_ in loaders[mode]]
        got = torch.cat(xs) if xs else torch.empty(0)
        assert got.shape[0] == ml[mode]
        if ml[mode]:
            assert torch.equal(got, x_ref[start:start + ml[mode]])
            ys = torch.cat([y for _, y, *_ in loaders[mode]])
            assert torch.equal(ys, y_ref[start:start + ml[mode]])
        start += ml[mode]


@settings(max_examples=20, deadline=None)
@given(obs=st.integers(2, 9), T=st.integers(40, 80), batch=st.integers(1, 5),
       mode=st.sampled_from(["train", "validate", "test"]))
def test_day_of_week_key_property(obs, T, batch, mode):
    """key = (obs + global_index) % 7 for EVERY sample of EVERY mode."""
    p = _params(N=6, T=T, batch=batch, obs=obs, pred=1)
    data = DataInput(p).load_data()
    gen = DataGenerator(obs, 1, p["split_ratio"])
    loaders = gen.get_data_loader(data, p)
    O_dyn = data["O_dyn_G"]
    n = data["OD"].shape[0] - obs - 1
    ml = gen.split2len(n)
    g = {"train": 0, "validate": ml["train"],
         "test": ml["train"] + ml["validate"]}[mode]
    for x, y, O_g, D_g in loaders[mode]:
        for i in range(O_g.shape[0]):
            assert torch.equal(O_g[i], O_dyn[:, :, (obs + g + i) % 7])
        g += O_g.shape[0]
