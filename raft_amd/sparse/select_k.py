"""select_k over CSR rows (reference: raft/sparse/matrix/detail/select_k-inl.cuh
— the dense radix/warpsort machinery through a CSR row-offset layout adapter).

Rows shorter than k pad with +/-inf; returned indices are COLUMN ids.
"""
from __future__ import annotations

import torch

from .types import CSR


def csr_select_k(a: CSR, k: int, select_min: bool = True):
    """Per-row top-k of CSR values. Returns (vals [n_rows,k], col_idx [n_rows,k]);
    missing slots hold +inf/-inf and index -1."""
    pad = float("inf") if select_min else float("-inf")
    lengths = (a.indptr[1:] - a.indptr[:-1]).to(torch.int64)
    max_len = int(lengths.max().item()) if lengths.numel() else 0
    width = max(max_len, k)
    dense = torch.full((a.n_rows, width), pad, dtype=a.values.dtype, device=a.device)
    cols = torch.full((a.n_rows, width), -1, dtype=torch.int64, device=a.device)
    seg = torch.repeat_interleave(torch.arange(a.n_rows, device=a.device), lengths)
    pos = torch.arange(a.nnz, device=a.device) - a.indptr[:-1].to(torch.int64)[seg]
    dense[seg, pos] = a.values
    cols[seg, pos] = a.indices.to(torch.int64)
    from raft_amd.matrix.select_k import select_k as dense_select_k
    vals, idx = dense_select_k(dense, k, select_min=select_min)
    return vals, torch.gather(cols, 1, idx)
