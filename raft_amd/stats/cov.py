"""Covariance (reference: raft/stats/cov.cuh — mean_center + syrk-style gemm)."""
from __future__ import annotations

import torch

from .moments import mean


def cov(x: torch.Tensor, sample: bool = True, centered: bool = False,
        weights: torch.Tensor | None = None) -> torch.Tensor:
    """Covariance matrix: mean-center + GEMM (reference stats::cov).

    weights: optional per-row FREQUENCY weights (numpy.cov fweights
    semantics — integer weights equal row repetition): weighted mean
    centering and denominator sum(w) - 1 (sample=True) or sum(w).
    """
    n = x.shape[0]
    if weights is None:
        xc = x if centered else x - mean(x).unsqueeze(0)
        return (xc.t() @ xc) / (n - 1 if sample else n)
    w = weights.to(x.dtype).clamp_min(0)
    sw = w.sum()
    mu = (w.unsqueeze(1) * x).sum(dim=0) / sw
    xc = x if centered else x - mu.unsqueeze(0)
    g = (xc * w.unsqueeze(1)).t() @ xc
    denom = sw - 1 if sample else sw           # fweights convention
    return g / denom
