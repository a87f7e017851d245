"""npz ingestion path: the reference's exact file contract
(od_day20180101_20210228.npz sparse, densified to (-1, 47, 47), last 425 days,
log1p; adjacency_matrix.npy) — Data_Container_OD.py:15-19,34."""

import numpy as np
import pytest
import scipy.sparse as ss
import torch

from mpgcn_amd.data import DataGenerator, DataInput


@pytest.fixture
def data_dir(tmp_path):
    T, N = 430, 47
    rng = np.random.default_rng(0)
    raw = rng.poisson(5.0, size=(T, N * N)).astype(np.float64)
    ss.save_npz(str(tmp_path / "od_day20180101_20210228.npz"),
                ss.csr_matrix(raw))
    np.save(str(tmp_path / "adjacency_matrix.npy"),
            rng.integers(0, 2, size=(N, N)).astype(np.float64))
    return tmp_path, raw


def test_npz_contract(data_dir):
    tmp_path, raw = data_dir
    params = {"input_dir": str(tmp_path), "norm": "none",
              "split_ratio": [6.4, 1.6, 2]}
    data = DataInput(params).load_data()
    OD = data["OD"]
    assert OD.shape == (425, 47, 47, 1)  # last 425 days, channel axis
    expect = np.log(raw[-425:].reshape(425, 47, 47) + 1.0)
    np.testing.assert_allclose(OD.squeeze(-1).numpy(), expect, rtol=1e-6)
    assert data["adj"].shape == (47, 47)
    assert data["O_dyn_G"].shape == (47, 47, 7)
    assert data["D_dyn_G"].shape == (47, 47, 7)


def test_npz_end_to_end_loader(data_dir):
    tmp_path, _ = data_dir
    params = {"input_dir": str(tmp_path), "norm": "minmax",
              "split_ratio": [6.4, 1.6, 2], "batch_size": 4,
              "obs_len": 7, "pred_len": 1, "seed": 0}
    data = DataInput(params).load_data()
    gen = DataGenerator(7, 1, params["split_ratio"])
    loaders = gen.get_data_loader(data, params)
    x, y, og, dg = next(iter(loaders["train"]))
    assert x.shape == (4, 7, 47, 47, 1)
    assert y.shape == (4, 1, 47, 47, 1)
    assert torch.isfinite(x).all()
