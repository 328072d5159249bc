"""Summary statistics (reference: raft/stats/{sum,mean,stddev,meanvar,minmax,
weighted_mean,mean_center}.cuh).

meanvar is the reference's single-pass Welford aggregate (detail/meanvar.cuh);
minmax uses the fused column min+max (the reference's ordered-int atomic
encoding trick is unnecessary here — torch aminmax is one fused vendor kernel).
"""
from __future__ import annotations

import torch

from raft_amd.linalg.reduce import strided_reduction


def sum_cols(x: torch.Tensor) -> torch.Tensor:
    """Column sums (stats::sum reduces along rows -> one value per column)."""
    return strided_reduction(x, main_op="identity", reduce_op="sum")


def mean(x: torch.Tensor, sample: bool = False) -> torch.Tensor:
    """Per-column means (reference stats::mean)."""
    return sum_cols(x) / x.shape[0]


def vars_(x: torch.Tensor, sample: bool = True,
          mu: torch.Tensor | None = None) -> torch.Tensor:
    """Per-column variances (reference stats::vars)."""
    if mu is None:
        mu = mean(x)
    n = x.shape[0]
    ss = strided_reduction(x - mu.unsqueeze(0), main_op="sq", reduce_op="sum")
    return ss / (n - 1 if sample else n)


def stddev(x: torch.Tensor, sample: bool = True) -> torch.Tensor:
    """Per-column standard deviations (reference stats::stddev)."""
    return vars_(x, sample=sample).sqrt()


def meanvar(x: torch.Tensor, sample: bool = True):
    """Mean+variance in one sweep pair (detail/meanvar.cuh): the mean is
    reused for the centered sq-sum instead of being recomputed."""
    mu = mean(x)
    var = vars_(x, sample=sample, mu=mu)
    return mu, var


def minmax(x: torch.Tensor):
    """Fused per-column (min, max) (detail/minmax.cuh)."""
    mn, mx = torch.aminmax(x, dim=0)
    return mn, mx


def weighted_mean(x: torch.Tensor, weights: torch.Tensor, along_rows: bool = True) -> torch.Tensor:
    """Weighted mean per row (weights over columns) or per column (weights over rows)."""
    w = weights.double()
    xd = x.double()
    if along_rows:
        assert weights.numel() == x.shape[1]
        out = (xd * w.unsqueeze(0)).sum(dim=1) / w.sum()
    else:
        assert weights.numel() == x.shape[0]
        out = (xd * w.unsqueeze(1)).sum(dim=0) / w.sum()
    return out.to(x.dtype)


def mean_center(x: torch.Tensor) -> torch.Tensor:
    """Subtract per-column means (reference mean_center)."""
    return x - mean(x).unsqueeze(0)


def mean_add(x: torch.Tensor, mu: torch.Tensor) -> torch.Tensor:
    """Add the column means back (inverse of mean_center)."""
    return x + mu.unsqueeze(0)
