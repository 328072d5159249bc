"""Covariance (reference: raft/stats/cov.cuh — mean_center + syrk-style gemm)."""
from __future__ import annotations

import torch

from .moments import mean


def cov(x: torch.Tensor, sample: bool = True, centered: bool = False) -> torch.Tensor:
    """Covariance matrix: mean-center + GEMM (reference stats::cov)."""
    xc = x if centered else x - mean(x).unsqueeze(0)
    n = x.shape[0]
    return (xc.t() @ xc) / (n - 1 if sample else n)
