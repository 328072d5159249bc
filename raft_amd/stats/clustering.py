"""Clustering quality metrics.

Reference parity: raft/stats/{adjusted_rand_index,rand_index,mutual_info_score,
entropy,homogeneity_score,completeness,v_measure,kl_divergence,dispersion,
silhouette_score}.cuh. silhouette is re-implemented against our own pairwise
distance (the reference's copy is vestigial post-cuVS-split).
"""
from __future__ import annotations

import math

import torch

from .contingency import contingency_matrix
from raft_amd.distance import pairwise_distance, DistanceType


def _comb2(x: torch.Tensor) -> torch.Tensor:
    x = x.double()
    return x * (x - 1.0) / 2.0


def adjusted_rand_index(a: torch.Tensor, b: torch.Tensor) -> float:
    """Adjusted Rand index between two labelings (reference ARI)."""
    c = contingency_matrix(a, b).double()
    n = c.sum()
    sum_comb = _comb2(c).sum()
    sum_a = _comb2(c.sum(dim=1)).sum()
    sum_b = _comb2(c.sum(dim=0)).sum()
    expected = sum_a * sum_b / _comb2(n)
    max_index = 0.5 * (sum_a + sum_b)
    if float(max_index - expected) == 0.0:
        return 1.0
    return float((sum_comb - expected) / (max_index - expected))


def rand_index(a: torch.Tensor, b: torch.Tensor) -> float:
    """O(1) via contingency (the reference's O(n^2/2) pair kernel is the
    brute-force formulation of the same count)."""
    c = contingency_matrix(a, b).double()
    n = c.sum()
    s = _comb2(c).sum()
    sa = _comb2(c.sum(dim=1)).sum()
    sb = _comb2(c.sum(dim=0)).sum()
    total = _comb2(n)
    return float((total + 2 * s - sa - sb) / total)


def mutual_info_score(a: torch.Tensor, b: torch.Tensor) -> float:
    """Mutual information between two labelings (nats)."""
    c = contingency_matrix(a, b).double()
    n = c.sum()
    p = c / n
    pa = p.sum(dim=1, keepdim=True)
    pb = p.sum(dim=0, keepdim=True)
    mask = p > 0
    terms = torch.where(mask, p * torch.log(p / (pa @ pb).clamp_min(1e-300)),
                        torch.zeros_like(p))
    return float(terms.sum())


def entropy(labels: torch.Tensor, n_classes: int | None = None) -> float:
    """Shannon entropy of a labeling (nats; reference stats::entropy)."""
    l = labels.to(torch.int64)
    counts = torch.bincount(l - int(l.min()), minlength=n_classes or 0).double()
    p = counts[counts > 0] / counts.sum()
    return float(-(p * p.log()).sum())


def homogeneity_score(truth: torch.Tensor, pred: torch.Tensor) -> float:
    """Homogeneity of pred w.r.t. truth (reference metric)."""
    h_c = entropy(truth)
    if h_c == 0.0:
        return 1.0
    mi = mutual_info_score(truth, pred)
    return mi / h_c


def completeness_score(truth: torch.Tensor, pred: torch.Tensor) -> float:
    """Completeness of pred w.r.t. truth (reference metric)."""
    return homogeneity_score(pred, truth)


def v_measure(truth: torch.Tensor, pred: torch.Tensor, beta: float = 1.0) -> float:
    """Harmonic mean of homogeneity and completeness (reference v_measure)."""
    h = homogeneity_score(truth, pred)
    c = completeness_score(truth, pred)
    if h + c == 0.0:
        return 0.0
    return float((1 + beta) * h * c / (beta * h + c))


def kl_divergence(p: torch.Tensor, q: torch.Tensor) -> float:
    """KL divergence between two distributions (reference metric)."""
    pd, qd = p.double(), q.double()
    mask = pd > 0
    return float(torch.where(mask, pd * (pd / qd.clamp_min(1e-300)).log(),
                             torch.zeros_like(pd)).sum())


def dispersion(x: torch.Tensor, labels: torch.Tensor, n_clusters: int) -> float:
    """Sum of squared distances of cluster centroids to the global centroid
    (reference: mean-dist-to-center reduction)."""
    labels = labels.to(torch.int64)
    d = x.shape[1]
    sums = torch.zeros((n_clusters, d), dtype=torch.float64, device=x.device)
    sums.index_add_(0, labels, x.double())
    counts = torch.bincount(labels, minlength=n_clusters).double().clamp_min(1)
    centroids = sums / counts.unsqueeze(1)
    global_c = x.double().mean(dim=0)
    return float((counts * ((centroids - global_c) ** 2).sum(dim=1)).sum().sqrt())


def silhouette_score(x: torch.Tensor, labels: torch.Tensor, n_clusters: int | None = None,
                     chunk: int = 4096) -> float:
    """Mean silhouette coefficient, chunked over rows (re-implemented on our
    pairwise distance; the reference's header is vestigial)."""
    labels = labels.to(torch.int64)
    k = n_clusters or int(labels.max().item()) + 1
    n = x.shape[0]
    counts = torch.bincount(labels, minlength=k).double()
    s_total = 0.0
    for s in range(0, n, chunk):
        e = min(s + chunk, n)
        d = pairwise_distance(x[s:e], x, metric=DistanceType.L2SqrtExpanded).double()
        # mean distance from each row to each cluster
        sums = torch.zeros((e - s, k), dtype=torch.float64, device=x.device)
        sums.index_add_(1, labels, d)
        own = labels[s:e]
        own_counts = counts[own]
        a = torch.where(own_counts > 1,
                        (sums.gather(1, own.unsqueeze(1)).squeeze(1)) / (own_counts - 1),
                        torch.zeros(e - s, dtype=torch.float64, device=x.device))
        meand = sums / counts.clamp_min(1).unsqueeze(0)
        meand.scatter_(1, own.unsqueeze(1), float("inf"))
        b = meand.min(dim=1).values
        sil = torch.where(own_counts > 1, (b - a) / torch.maximum(a, b),
                          torch.zeros_like(a))
        s_total += float(sil.sum())
    return s_total / n


def silhouette_score_batched(x: torch.Tensor, labels: torch.Tensor,
                             n_clusters: int | None = None,
                             batch_size: int = 4096) -> float:
    """Batched silhouette (reference stats/detail/batched/silhouette_score):
    identical result to silhouette_score, with the row-batch size exposed so
    the [batch, n] distance block is bounded on huge inputs."""
    return silhouette_score(x, labels, n_clusters=n_clusters, chunk=batch_size)
