"""COO/CSR structural ops (reference: raft/sparse/op/* — sort, filter,
dedupe, row_op, slice)."""
from __future__ import annotations

import torch

from .types import CSR, COO


def coo_sort(coo: COO) -> COO:
    """Sort COO triplets by (row, col) (reference coo sort)."""
    order = torch.argsort(coo.rows.to(torch.int64) * coo.n_cols + coo.cols.to(torch.int64))
    return COO(coo.rows[order], coo.cols[order], coo.values[order],
               coo.n_rows, coo.n_cols)


def filter_zeros(coo: COO, eps: float = 0.0) -> COO:
    """Remove entries with |v| <= eps (op/filter remove-zeros)."""
    keep = coo.values.abs() > eps
    return COO(coo.rows[keep], coo.cols[keep], coo.values[keep],
               coo.n_rows, coo.n_cols)


def dedupe_coo(coo: COO, op: str = "max") -> COO:
    """Coalesce duplicate coordinates (op/reduce.cuh compute_duplicates_mask)."""
    key = coo.rows.to(torch.int64) * coo.n_cols + coo.cols.to(torch.int64)
    uniq, inv = torch.unique(key, return_inverse=True)
    out = torch.zeros(uniq.numel(), dtype=coo.values.dtype, device=coo.device)
    if op == "max":
        out.fill_(float("-inf"))
        out.scatter_reduce_(0, inv, coo.values, reduce="amax")
    elif op == "add":
        out.index_add_(0, inv, coo.values)
    else:
        raise ValueError(op)
    return COO(uniq // coo.n_cols, uniq % coo.n_cols, out, coo.n_rows, coo.n_cols)


def slice_csr_rows(a: CSR, start: int, stop: int) -> CSR:
    """Row-range slice of a CSR matrix (reference slice csr rows)."""
    lo = int(a.indptr[start].item())
    hi = int(a.indptr[stop].item())
    indptr = a.indptr[start:stop + 1] - lo
    return CSR(indptr, a.indices[lo:hi].clone(), a.values[lo:hi].clone(),
               stop - start, a.n_cols)


def csr_row_op(a: CSR, fn) -> CSR:
    """Apply fn(row_values) -> new values per row (op/row_op.cuh)."""
    lengths = (a.indptr[1:] - a.indptr[:-1]).to(torch.int64)
    seg = torch.repeat_interleave(torch.arange(a.n_rows, device=a.device), lengths)
    new_vals = fn(a.values, seg)
    return CSR(a.indptr, a.indices, new_vals, a.n_rows, a.n_cols)


def csr_diagonal(a: CSR) -> torch.Tensor:
    """Extract the main diagonal (reference: sparse/matrix/diagonal.cuh).
    Missing diagonal entries read as 0."""
    out = torch.zeros(min(a.n_rows, a.n_cols), dtype=a.values.dtype,
                      device=a.values.device)
    lengths = a.indptr[1:] - a.indptr[:-1]
    rows = torch.repeat_interleave(
        torch.arange(a.n_rows, device=a.values.device), lengths.long())
    on_diag = a.indices.long() == rows
    out[rows[on_diag]] = a.values[on_diag]
    return out


def csr_set_diagonal(a: CSR, vec: torch.Tensor) -> CSR:
    """Set existing diagonal entries to vec[i] (reference diagonal.cuh
    set_diagonal: only stored positions are written — the sparsity pattern
    is unchanged)."""
    lengths = a.indptr[1:] - a.indptr[:-1]
    rows = torch.repeat_interleave(
        torch.arange(a.n_rows, device=a.values.device), lengths.long())
    on_diag = a.indices.long() == rows
    values = a.values.clone()
    values[on_diag] = vec.to(values.dtype)[rows[on_diag]]
    return CSR(a.indptr, a.indices, values, a.n_rows, a.n_cols)
