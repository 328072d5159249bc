"""select_k over CSR rows (reference: raft/sparse/matrix/detail/select_k-inl.cuh
— the dense radix/warpsort machinery through a CSR row-offset layout adapter).

MI355X design: on GPU the native generic select kernel streams each CSR row
through its row-offset window directly (csrc/select_k.hip
select_k_generic_kernel) — NO densification (the round-1 version scattered
into an [n_rows x max_row_len] scratch, which blows up on power-law degree
distributions). CPU keeps the dense scatter (test oracle).

Rows shorter than k pad with +/-inf; returned indices are COLUMN ids, -1 for
padded slots.
"""
from __future__ import annotations

import torch

from .types import CSR


def csr_select_k(a: CSR, k: int, select_min: bool = True):
    """Per-row top-k of CSR values. Returns (vals [n_rows,k], col_idx [n_rows,k]);
    missing slots hold +inf/-inf and index -1."""
    if a.values.is_cuda and a.values.dtype in (torch.float32, torch.float64,
                                               torch.bfloat16, torch.float16):
        from raft_amd._ext import require_ext
        ext = require_ext()
        row_off = a.indptr.to(torch.int64).contiguous()
        vals, pos = ext.select_k_generic(a.values.contiguous(), row_off,
                                         k=int(k), select_min=bool(select_min))
        # pos is the within-row nnz position; map to column ids (-1 pads)
        valid = pos >= 0
        flat = (row_off[:-1].unsqueeze(1) + pos.clamp_min(0)).reshape(-1)
        cols = a.indices.to(torch.int64)[flat].reshape(pos.shape)
        cols = torch.where(valid, cols, torch.full_like(cols, -1))
        # sort each row (generic kernel output is unsorted)
        order = torch.argsort(torch.where(valid, vals,
                                          torch.full_like(vals, float("inf")
                                                          if select_min else
                                                          float("-inf"))),
                              dim=1, descending=not select_min)
        return torch.gather(vals, 1, order), torch.gather(cols, 1, order)
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
