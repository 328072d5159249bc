"""Neighborhood metrics.

Reference parity: raft/stats/neighborhood_recall.cuh (ANN ground-truth overlap)
and trustworthiness_score.cuh (re-implemented against our knn — the
reference's copy is vestigial post-cuVS-split).
"""
from __future__ import annotations

import torch

from raft_amd.neighbors.brute_force import knn
from raft_amd.distance import DistanceType


def neighborhood_recall(found_idx: torch.Tensor, truth_idx: torch.Tensor) -> float:
    """Mean fraction of true k-NN recovered, per row."""
    n, k = truth_idx.shape
    hits = 0
    f = found_idx.to(torch.int64)
    t = truth_idx.to(torch.int64)
    match = (f.unsqueeze(2) == t.unsqueeze(1)).any(dim=2)
    return float(match.double().mean())


def trustworthiness_score(x: torch.Tensor, x_embedded: torch.Tensor,
                          n_neighbors: int = 5) -> float:
    """Trustworthiness of an embedding (standard formula, exact knn ranks)."""
    n = x.shape[0]
    k = n_neighbors
    # ranks in original space
    _, emb_nn = knn(x_embedded, x_embedded, k + 1, metric=DistanceType.L2Expanded)
    emb_nn = emb_nn[:, 1:]  # drop self
    from raft_amd.distance import pairwise_distance
    d_orig = pairwise_distance(x, x, metric=DistanceType.L2Expanded)
    ranks = d_orig.argsort(dim=1).argsort(dim=1)  # rank of each point per row
    # self occupies rank 0, so the full-list rank of a non-self point IS its
    # 1-based rank among non-self points — the standard formula's r(i, j)
    r = ranks.gather(1, emb_nn)
    penalty = (r - k).clamp_min(0).double().sum()
    norm = n * k * (2.0 * n - 3.0 * k - 1.0)
    return float(1.0 - 2.0 / norm * penalty)
