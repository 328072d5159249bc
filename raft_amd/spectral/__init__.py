"""Spectral graph analysis (reference: raft/spectral/* — partition via
Laplacian eigenvectors + kmeans, analyzePartition edge-cut quadratic form,
analyzeModularity; matrix_wrappers.hpp sparse/laplacian/modularity operators).
"""
from .partition import partition, analyze_partition, analyze_modularity

__all__ = ["partition", "analyze_partition", "analyze_modularity"]
