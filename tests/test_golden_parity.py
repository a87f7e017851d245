"""Golden parity artifact (VERDICT round-1 item 8).

Builds a 47-region synthetic dataset in the reference's EXACT on-disk layout
(sparse `od_day20180101_20210228.npz` + `adjacency_matrix.npy`,
Data_Container_OD.py:15-17,34), loads it through the npz path with
`-ref-quirks` (reproducing the reference's D-graph column/row mixing,
Data_Container_OD.py:53-56), and asserts that the transcription oracle
(models/reference_eager.py) and the native stack produce IDENTICAL
scores-file lines from the same checkpoint — the full-stack bit-parity check
the round-1 verdict called the "last bit-parity door".

Both stacks run in float64 so the only differences left are algorithmic
(factored vs K^2-pair formulation); at f64 those round away below the 10
printed decimals.
"""

import numpy as np
import pytest
import scipy.sparse as ss
import scipy.spatial.distance as sdist
import torch

from mpgcn_amd.data import DataGenerator, DataInput
from mpgcn_amd.graph.dynamic import construct_dynamic_graphs
from mpgcn_amd.models import MPGCN
from mpgcn_amd.models.reference_eager import MPGCNReference
from mpgcn_amd.train import metrics as metrics_mod


def _write_reference_layout_dataset(tmp_path, N=47, days=425, seed=0):
    """The reference loader's exact file contract: one scipy sparse matrix
    whose densified form reshapes to (-1, N, N), and a dense (N, N) npy."""
    rng = np.random.default_rng(seed)
    raw = rng.poisson(4.0, size=(days, N, N)).astype(np.float64)
    raw *= rng.random((days, N, N)) < 0.7  # sparsity like real OD flows
    sp = ss.csr_matrix(raw.reshape(days * N, N))
    ss.save_npz(str(tmp_path / "od_day20180101_20210228.npz"), sp)
    adj = (rng.random((N, N)) < 0.25).astype(np.float64)
    np.save(str(tmp_path / "adjacency_matrix.npy"), adj)
    return raw, adj


def test_npz_reference_layout_roundtrip(tmp_path):
    raw, adj = _write_reference_layout_dataset(tmp_path)
    params = {"input_dir": str(tmp_path), "norm": "none",
              "split_ratio": [6.4, 1.6, 2], "ref_quirks": True}
    data = DataInput(params).load_data()
    assert data["OD"].shape == (425, 47, 47, 1)
    assert torch.allclose(data["OD"][..., 0].double(),
                          torch.from_numpy(np.log(raw + 1.0)), atol=1e-6)
    assert data["O_dyn_G"].shape == (47, 47, 7)
    assert data["D_dyn_G"].shape == (47, 47, 7)


def test_ref_quirks_matches_scipy_loop():
    """ref_quirks=True reproduces Data_Container_OD.py:45-56 verbatim,
    including the D-graph's column-i/row-j mixing, against a direct scipy
    transcription on f64 data."""
    rng = np.random.default_rng(3)
    period, N, weeks = 7, 12, 4
    hist = rng.random((weeks * period, N, N)) * 5 + 0.1
    O_dyn, D_dyn = construct_dynamic_graphs(
        torch.from_numpy(hist), period=period, ref_quirks=True
    )
    for t in range(period):
        avg = hist[t::period].mean(axis=0)
        for i in range(N):
            for j in range(N):
                o_ref = sdist.cosine(avg[i, :], avg[j, :])
                d_ref = sdist.cosine(avg[:, i], avg[j, :])  # the quirk
                assert abs(O_dyn[i, j, t].item() - o_ref) < 1e-10
                assert abs(D_dyn[i, j, t].item() - d_ref) < 1e-10


@pytest.mark.timeout(600)
def test_oracle_and_native_scores_lines_identical(tmp_path):
    from mpgcn_amd.graph import build_supports

    _write_reference_layout_dataset(tmp_path)
    params = {"input_dir": str(tmp_path), "norm": "none",
              "split_ratio": [6.4, 1.6, 2], "ref_quirks": True,
              "batch_size": 8, "seed": 0}
    data = DataInput(params).load_data()
    N, H, K_ord, pred_len = 47, 16, 2, 2
    kernel = "random_walk_diffusion"

    gen = DataGenerator(obs_len=7, pred_len=pred_len,
                        data_split_ratio=params["split_ratio"])
    loaders = gen.get_data_loader(data, params)

    torch.manual_seed(11)
    native = MPGCN(M=2, K=K_ord + 1, input_dim=1, lstm_hidden_dim=H,
                   lstm_num_layers=1, gcn_hidden_dim=H, gcn_num_layers=3,
                   num_nodes=N, compute_dtype=torch.float64).double()
    ckpt = {"epoch": 1, "state_dict": native.state_dict()}

    oracle = MPGCNReference(M=2, K=K_ord + 1, input_dim=1, hidden=H,
                            gcn_layers=3, num_nodes=N).double()
    oracle.load_state_dict(ckpt["state_dict"])  # same checkpoint, key-for-key

    # shared f64 graph preprocessing (one build, both stacks consume it)
    G_static = build_supports(
        data["adj"].double().unsqueeze(0), kernel, K_ord
    ).squeeze(0)

    def line_for(model):
        forecast, truth = [], []
        with torch.no_grad():
            for x, y, O_raw, D_raw in loaders["test"]:
                dyn = (build_supports(O_raw.double(), kernel, K_ord),
                       build_supports(D_raw.double(), kernel, K_ord))
                cur = x.double()
                preds = []
                for _ in range(pred_len):
                    step = model(cur, [G_static, dyn])
                    cur = torch.cat([cur[:, 1:], step], dim=1)
                    preds.append(step)
                forecast.append(torch.cat(preds, dim=1).numpy())
                truth.append(y.double().numpy())
        f = np.concatenate(forecast, axis=0)
        t = np.concatenate(truth, axis=0)
        return "test, MSE, RMSE, MAE, MAPE, %.10f, %.10f, %.10f, %.10f" % (
            metrics_mod.MSE(f, t), metrics_mod.RMSE(f, t),
            metrics_mod.MAE(f, t), metrics_mod.MAPE(f, t),
        )

    native_line = line_for(native)
    oracle_line = line_for(oracle)
    assert native_line == oracle_line, (native_line, oracle_line)
