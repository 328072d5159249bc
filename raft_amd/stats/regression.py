"""Regression metrics (reference: raft/stats/scores.cuh r2_score,
regression_metrics (MAE/MSE/MedAE via sort+median), information_criterion.cuh
(AIC/AICc/BIC))."""
from __future__ import annotations

import math

import torch


def r2_score(y_true: torch.Tensor, y_pred: torch.Tensor) -> float:
    """Coefficient of determination (reference r2_score)."""
    yt, yp = y_true.double(), y_pred.double()
    ss_res = ((yt - yp) ** 2).sum()
    ss_tot = ((yt - yt.mean()) ** 2).sum()
    return float(1.0 - ss_res / ss_tot.clamp_min(1e-300))


def regression_metrics(y_true: torch.Tensor, y_pred: torch.Tensor):
    """Returns (mean_abs_error, mean_squared_error, median_abs_error)."""
    err = (y_true.double() - y_pred.double()).abs()
    mae = float(err.mean())
    mse = float((err ** 2).mean())
    medae = float(err.median())
    return mae, mse, medae


def information_criterion(log_likelihood: float, n_params: int, n_samples: int,
                          kind: str = "aic") -> float:
    """AIC/AICc/BIC from log-likelihood (reference information_criterion)."""
    ll, k, n = float(log_likelihood), int(n_params), int(n_samples)
    if kind == "aic":
        return -2.0 * ll + 2.0 * k
    if kind == "aicc":
        return -2.0 * ll + 2.0 * k + (2.0 * k * (k + 1)) / max(n - k - 1, 1)
    if kind == "bic":
        return -2.0 * ll + k * math.log(n)
    raise ValueError(kind)
