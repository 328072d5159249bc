"""Sparse formats, conversions, linalg and solvers (reference: raft/sparse/*,
SURVEY §2.4). COO/CSR containers are thin records over torch tensors; vendor
ops go through torch.sparse (rocSPARSE on ROCm); the hot CSR SpMV used by
Lanczos (BASELINE config 4) is a hand-written wave64 HIP kernel.
"""
from .types import CSR, COO
from .convert import (
    coo_to_csr, csr_to_coo, csr_to_dense, dense_to_csr, adj_to_csr,
    bitmap_to_csr, sorted_coo_to_csr,
)
from .linalg import (
    spmv, spmm, sddmm, masked_matmul, laplacian, laplacian_normalized,
    symmetrize_coo, knn_graph_symmetrize, csr_transpose, csr_row_norm,
    csr_degree, csr_add,
)
from .op import (coo_sort, filter_zeros, dedupe_coo, slice_csr_rows, csr_row_op,
                 csr_diagonal, csr_set_diagonal)
from .select_k import csr_select_k
from .preprocessing import tfidf_transform, bm25_transform
from . import solver

__all__ = [
    "CSR", "COO", "coo_to_csr", "csr_to_coo", "csr_to_dense", "dense_to_csr",
    "adj_to_csr", "bitmap_to_csr", "sorted_coo_to_csr",
    "spmv", "spmm", "sddmm", "masked_matmul", "laplacian", "laplacian_normalized",
    "symmetrize_coo", "knn_graph_symmetrize", "csr_transpose", "csr_row_norm",
    "csr_degree", "csr_add",
    "coo_sort", "filter_zeros", "dedupe_coo", "slice_csr_rows", "csr_row_op",
    "csr_select_k", "tfidf_transform", "bm25_transform", "solver",
]
