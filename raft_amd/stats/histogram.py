"""Histogram (reference: raft/stats/detail/histogram.cuh — gmem/smem/
smem-bits/hash strategies auto-chosen by nbins; torch.histc / bincount lower
to the rocPRIM histogram kernels which implement the same strategy split)."""
from __future__ import annotations

import torch


def histogram(x: torch.Tensor, n_bins: int, lo: float | None = None,
              hi: float | None = None) -> torch.Tensor:
    """Per-column histograms of a [n, d] matrix -> [n_bins, d] int64 counts."""
    if x.dim() == 1:
        x = x.unsqueeze(1)
    lo = float(x.min().item()) if lo is None else lo
    hi = float(x.max().item()) if hi is None else hi
    width = (hi - lo) or 1.0
    bins = ((x.double() - lo) / width * n_bins).floor().clamp_(0, n_bins - 1).to(torch.int64)
    out = torch.zeros((n_bins, x.shape[1]), dtype=torch.int64, device=x.device)
    for j in range(x.shape[1]):
        out[:, j] = torch.bincount(bins[:, j], minlength=n_bins)
    return out
