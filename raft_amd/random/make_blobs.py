"""Gaussian-cluster synthetic data generator.

Reference parity: raft/random/make_blobs.cuh:58,126 + detail/make_blobs.cuh:88
(single fused generate_data_kernel: sample cluster id + centroid offset per
point in one pass).

MI355X: the GPU path is one fused HIP kernel (csrc/rng.hip make_blobs) —
cluster assignment, Box-Muller noise, centroid add and label write in a single
HBM-write pass; centroids stream from L2 (k*d*4 bytes is tiny).
"""
from __future__ import annotations

import torch

from raft_amd._ext import require_ext
from .rng import RngState, normal, uniform, uniform_int


def make_blobs(n_rows: int, n_cols: int, n_clusters: int = 5,
               cluster_std: float = 1.0, centers: torch.Tensor | None = None,
               center_box: tuple = (-10.0, 10.0), shuffle: bool = True,
               state: RngState | None = None, device=None, dtype=torch.float32):
    """Returns (X [n_rows, n_cols], labels [n_rows] int64, centers [k, n_cols])."""
    state = state or RngState(seed=0)
    device = torch.device(device) if device is not None else torch.device("cpu")
    if centers is None:
        centers = uniform((n_clusters, n_cols), center_box[0], center_box[1],
                          state=state, device=device, dtype=torch.float32)
    else:
        centers = centers.to(device=device, dtype=torch.float32)
        n_clusters = centers.shape[0]

    if device.type == "cuda":
        ext = require_ext()
        x, labels = ext.make_blobs(int(n_rows), int(n_cols), centers.contiguous(),
                                   float(cluster_std), int(state.seed),
                                   int(state.base_subsequence))
        state.advance(3)
        return x.to(dtype), labels.to(torch.int64), centers

    labels = uniform_int((n_rows,), 0, n_clusters, state=state, device=device)
    noise = normal((n_rows, n_cols), 0.0, cluster_std, state=state, device=device,
                   dtype=torch.float32)
    x = centers[labels] + noise
    return x.to(dtype), labels, centers
