"""Spectral partitioning and its quality metrics.

Reference parity: raft/spectral/partition.cuh:38 (Laplacian eigensolve via
Lanczos -> kmeans on the eigenvector embedding), detail/partition.hpp:47-90
(analyzePartition: per-cluster indicator vector + Laplacian quadratic form
edge cut), modularity_maximization.cuh:31.
"""
from __future__ import annotations

import torch

from raft_amd.sparse.types import CSR
from raft_amd.sparse.linalg import laplacian, spmv, csr_degree
from raft_amd.sparse.solver.lanczos import lanczos_min_eigenpairs, LanczosConfig
from raft_amd.cluster.kmeans import kmeans_fit, KMeansParams


def partition(a: CSR, n_clusters: int, n_eig: int | None = None,
              seed: int = 42, tol: float = 1e-6):
    """Spectral partition of graph `a`. Returns (labels, eigenvalues, eigenvectors)."""
    n_eig = n_eig or n_clusters
    lap = laplacian(a)
    cfg = LanczosConfig(n_components=n_eig, tolerance=tol, seed=seed)
    w, v = lanczos_min_eigenpairs(lap, k=n_eig, config=cfg)
    emb = v.to(torch.float32).contiguous()
    model = kmeans_fit(emb, KMeansParams(n_clusters=n_clusters, max_iter=50,
                                         seed=seed, init="kmeans++"))
    from raft_amd.neighbors.fused_l2nn import fused_l2nn
    labels = fused_l2nn(emb, model.centroids)[1]
    return labels, w, v


def analyze_partition(a: CSR, labels: torch.Tensor, n_clusters: int | None = None):
    """Edge cut + cluster sizes via the Laplacian quadratic form
    (detail/partition.hpp:47-90): cut = 1/2 sum_k x_k^T L x_k."""
    k = n_clusters or int(labels.max().item()) + 1
    lap = laplacian(a)
    labels = labels.to(torch.int64)
    edge_cut = 0.0
    sizes = []
    for c in range(k):
        x = (labels == c).to(a.values.dtype)
        sizes.append(int(x.sum().item()))
        edge_cut += 0.5 * float(torch.dot(x, spmv(lap, x)).item())
    return edge_cut, sizes


def analyze_modularity(a: CSR, labels: torch.Tensor, n_clusters: int | None = None) -> float:
    """Newman modularity Q = sum_k (e_kk/m - (d_k/2m)^2) (modularity_maximization.cuh)."""
    k = n_clusters or int(labels.max().item()) + 1
    labels = labels.to(torch.int64)
    deg = torch.zeros(a.n_rows, dtype=torch.float64, device=a.device)
    seg = torch.repeat_interleave(torch.arange(a.n_rows, device=a.device),
                                  (a.indptr[1:] - a.indptr[:-1]).to(torch.int64))
    deg.index_add_(0, seg, a.values.double())
    two_m = float(deg.sum().item())
    q = 0.0
    for c in range(k):
        x = (labels == c).to(a.values.dtype)
        e_cc = float(torch.dot(x, spmv(a, x)).item())
        d_c = float(deg[labels == c].sum().item())
        q += e_cc / two_m - (d_c / two_m) ** 2
    return q
