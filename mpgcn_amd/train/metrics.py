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
