"""Evaluation metrics — definition-compatible with the reference Metrics.py:
MSE, RMSE, MAE, MAPE with epsilon = 1.0 in the denominator (Metrics.py:22-23),
PCC (Pearson on flattened arrays; printed but not returned by evaluate,
Metrics.py:5-11). Numpy API mirrors the reference; torch variants run the same
reductions on device for large-N evaluation without a host round trip.
"""

from __future__ import annotations

import numpy as np
import torch


def MSE(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.mean(np.square(y_pred - y_true)))


def RMSE(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.sqrt(MSE(y_pred, y_true)))


def MAE(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.mean(np.abs(y_pred - y_true)))


def MAPE(y_pred: np.ndarray, y_true: np.ndarray, epsilon: float = 1e-0) -> float:
    return float(np.mean(np.abs(y_pred - y_true) / (y_true + epsilon)))


def PCC(y_pred: np.ndarray, y_true: np.ndarray) -> float:
    return float(np.corrcoef(y_pred.flatten(), y_true.flatten())[0, 1])


def evaluate(y_pred: np.ndarray, y_true: np.ndarray, precision: int = 4):
    """Prints all five metrics, returns (MSE, RMSE, MAE, MAPE) — the exact
    contract of Metrics.py:5-11."""
    print("MSE:", round(MSE(y_pred, y_true), precision))
    print("RMSE:", round(RMSE(y_pred, y_true), precision))
    print("MAE:", round(MAE(y_pred, y_true), precision))
    print("MAPE:", round(MAPE(y_pred, y_true) * 100, precision), "%")
    print("PCC:", round(PCC(y_pred, y_true), precision))
    return (
        MSE(y_pred, y_true),
        RMSE(y_pred, y_true),
        MAE(y_pred, y_true),
        MAPE(y_pred, y_true),
    )


class MetricAccumulator:
    """Streaming sufficient statistics for (MSE, RMSE, MAE, MAPE, PCC) —
    update per batch on device, optionally all-reduce across ranks, finalize
    once. Lets evaluation run sharded (each rank sees a subset of batches or a
    destination shard of each batch) and in O(1) memory instead of
    concatenating the full forecast tensor (cf. the reference's host
    accumulation, Model_Trainer.py:169-173). f64 accumulation matches the
    numpy path to ~1e-12; cross-rank summation order can move the last
    printed digit."""

    def __init__(self, device="cpu"):
        # [n, sum_se, sum_ae, sum_ape, sum_p, sum_t, sum_pt, sum_p2, sum_t2]
        self.s = torch.zeros(9, dtype=torch.float64, device=device)

    @torch.no_grad()
    def update(self, y_pred: torch.Tensor, y_true: torch.Tensor):
        p = y_pred.double().flatten()
        t = y_true.double().flatten()
        d = p - t
        self.s += torch.stack([
            torch.tensor(float(p.numel()), dtype=torch.float64, device=p.device),
            d.square().sum(), d.abs().sum(), (d.abs() / (t + 1.0)).sum(),
            p.sum(), t.sum(), (p * t).sum(), p.square().sum(), t.square().sum(),
        ])

    def all_reduce(self, group=None):
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(self.s, group=group)

    def finalize(self):
        n, sse, sae, sape, sp, st, spt, sp2, st2 = self.s.cpu().tolist()
        if n == 0:
            return (float("nan"),) * 5
        mse = sse / n
        pm, tm = sp / n, st / n
        cov = spt / n - pm * tm
        var_p = max(sp2 / n - pm * pm, 0.0)
        var_t = max(st2 / n - tm * tm, 0.0)
        pcc = cov / ((var_p * var_t) ** 0.5 + 1e-300)
        return (mse, mse ** 0.5, sae / n, sape / n, pcc)


@torch.no_grad()
def evaluate_torch(y_pred: torch.Tensor, y_true: torch.Tensor):
    """Device-side (MSE, RMSE, MAE, MAPE, PCC) as a 5-tuple of floats."""
    y_pred = y_pred.float()
    y_true = y_true.float()
    diff = y_pred - y_true
    mse = diff.square().mean()
    mae = diff.abs().mean()
    mape = (diff.abs() / (y_true + 1.0)).mean()
    p = y_pred.flatten()
    t = y_true.flatten()
    pm, tm = p.mean(), t.mean()
    cov = ((p - pm) * (t - tm)).mean()
    pcc = cov / (p.std(unbiased=False) * t.std(unbiased=False))
    return (
        mse.item(),
        mse.sqrt().item(),
        mae.item(),
        mape.item(),
        pcc.item(),
    )
