"""Histogram (reference: raft/stats/detail/histogram.cuh — gmem/smem/
smem-bits/hash strategies auto-chosen by nbins, histogram.cuh:52-85).

MI355X design: native kernel (csrc/histogram.hip) with wave-private LDS
sub-histograms for nbins <= 2048 (the wave64-native answer to the
reference's packed-counter contention strategy), single-LDS to 16384 bins,
gmem atomics beyond; grid = (column, row-chunk). The round-1 per-column
torch.bincount Python loop (d kernel launches + host loop) is the CPU
fallback/oracle.
"""
from __future__ import annotations

import torch


def histogram(x: torch.Tensor, n_bins: int, lo: float | None = None,
              hi: float | None = None) -> torch.Tensor:
    """Per-column histograms of a [n, d] matrix -> [n_bins, d] int64 counts."""
    if x.dim() == 1:
        x = x.unsqueeze(1)
    lo = float(x.min().item()) if lo is None else lo
    hi = float(x.max().item()) if hi is None else hi
    if x.is_cuda and x.dtype == torch.float32:
        from raft_amd._ext import require_ext
        return require_ext().histogram_f32(x.contiguous(), int(n_bins), lo, hi)
    width = (hi - lo) or 1.0
    bins = ((x.double() - lo) / width * n_bins).floor().clamp_(0, n_bins - 1).to(torch.int64)
    out = torch.zeros((n_bins, x.shape[1]), dtype=torch.int64, device=x.device)
    for j in range(x.shape[1]):
        out[:, j] = torch.bincount(bins[:, j], minlength=n_bins)
    return out
