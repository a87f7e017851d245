"""Metrics vs reference definitions (Metrics.py:5-26), incl. MAPE eps=1.0."""

import numpy as np
import torch

from mpgcn_amd.train import metrics as M


def test_values():
    rng = np.random.default_rng(0)
    p = rng.normal(size=(4, 5, 5)).astype(np.float64)
    t = rng.normal(size=(4, 5, 5)).astype(np.float64) + 2.0
    assert np.isclose(M.MSE(p, t), np.mean((p - t) ** 2))
    assert np.isclose(M.RMSE(p, t), np.sqrt(np.mean((p - t) ** 2)))
    assert np.isclose(M.MAE(p, t), np.mean(np.abs(p - t)))
    assert np.isclose(M.MAPE(p, t), np.mean(np.abs(p - t) / (t + 1.0)))
    assert np.isclose(M.PCC(p, t), np.corrcoef(p.flatten(), t.flatten())[0, 1])


def test_evaluate_returns_four(capsys):
    p = np.ones((3, 3))
    t = np.full((3, 3), 2.0)
    res = M.evaluate(p, t)
    assert len(res) == 4
    out = capsys.readouterr().out
    assert "PCC:" in out  # printed but not returned (Metrics.py:10-11)


def test_torch_matches_numpy():
    rng = np.random.default_rng(1)
    p = rng.normal(size=(100,)).astype(np.float32)
    t = rng.normal(size=(100,)).astype(np.float32) + 3.0
    mse, rmse, mae, mape, pcc = M.evaluate_torch(torch.from_numpy(p), torch.from_numpy(t))
    assert np.isclose(mse, M.MSE(p, t), rtol=1e-5)
    assert np.isclose(rmse, M.RMSE(p, t), rtol=1e-5)
    assert np.isclose(mae, M.MAE(p, t), rtol=1e-5)
    assert np.isclose(mape, M.MAPE(p, t), rtol=1e-5)
    assert np.isclose(pcc, M.PCC(p, t), rtol=1e-4)


def test_evaluate_torch_matches_numpy():
    """Device-side metric variants agree with the reference-contract numpy
    definitions (Metrics.py:5-26) on random data."""
    import numpy as np
    import torch

    from mpgcn_amd.train import metrics as m

    rng = np.random.default_rng(0)
    yp = rng.random((50, 7, 9, 9, 1)).astype(np.float32) * 3
    yt = rng.random((50, 7, 9, 9, 1)).astype(np.float32) * 3
    mse, rmse, mae, mape = m.evaluate(yp, yt)
    tmse, trmse, tmae, tmape, tpcc = m.evaluate_torch(
        torch.from_numpy(yp), torch.from_numpy(yt))
    assert abs(mse - tmse) < 1e-5
    assert abs(rmse - trmse) < 1e-5
    assert abs(mae - tmae) < 1e-6
    assert abs(mape - tmape) < 1e-6
    assert abs(m.PCC(yp, yt) - tpcc) < 1e-5


# ---- property sweep: MetricAccumulator == whole-array numpy on any split ----
from hypothesis import given, settings, strategies as st  # noqa: E402


@settings(max_examples=30, deadline=None)
@given(n=st.integers(2, 400), chunks=st.integers(1, 6),
       scale=st.sampled_from([0.1, 1.0, 50.0]), seed=st.integers(0, 999))
def test_metric_accumulator_chunking_invariance(n, chunks, scale, seed):
    """Streaming sufficient statistics must be invariant to HOW the data is
    chunked and match the reference numpy definitions on the concatenation —
    this is exactly what distributed eval relies on (per-rank partial batches
    all-reduced, trainer.test)."""
    from mpgcn_amd.train.metrics import MetricAccumulator

    rng = np.random.default_rng(seed)
    p = (rng.normal(size=(n,)) * scale).astype(np.float64)
    t = (rng.normal(size=(n,)) * scale + 2.0).astype(np.float64)
    acc = MetricAccumulator()
    bounds = sorted(rng.integers(0, n, size=max(chunks - 1, 0)).tolist())
    pieces = np.split(np.arange(n), bounds)
    for idx in pieces:
        if len(idx):
            acc.update(torch.from_numpy(p[idx]), torch.from_numpy(t[idx]))
    mse, rmse, mae, mape, pcc = acc.finalize()
    assert np.isclose(mse, M.MSE(p, t), rtol=1e-10)
    assert np.isclose(rmse, M.RMSE(p, t), rtol=1e-10)
    assert np.isclose(mae, M.MAE(p, t), rtol=1e-10)
    assert np.isclose(mape, M.MAPE(p, t), rtol=1e-10)
    if np.std(p) > 1e-12 and np.std(t) > 1e-12:
        assert np.isclose(pcc, M.PCC(p, t), rtol=1e-8)
