"""Format conversions (reference: raft/sparse/convert/* — COO<->CSR, CSR<->dense,
adj(bool dense)->CSR, bitmap->CSR, sorted_coo_to_csr)."""
from __future__ import annotations

import torch

from raft_amd.core.bitset import Bitset
from .types import CSR, COO


def coo_to_csr(coo: COO) -> CSR:
    """COO -> CSR conversion (sorts by row; reference convert/csr)."""
    order = torch.argsort(coo.rows * coo.n_cols + coo.cols)
    rows = coo.rows[order]
    cols = coo.cols[order]
    vals = coo.values[order]
    counts = torch.bincount(rows.to(torch.int64), minlength=coo.n_rows)
    indptr = torch.zeros(coo.n_rows + 1, dtype=torch.int64, device=coo.device)
    torch.cumsum(counts, dim=0, out=indptr[1:])
    return CSR(indptr, cols, vals, coo.n_rows, coo.n_cols)


def sorted_coo_to_csr(coo: COO) -> CSR:
    """Row-sorted COO -> CSR by histogram only (convert/csr.cuh:48-63)."""
    counts = torch.bincount(coo.rows.to(torch.int64), minlength=coo.n_rows)
    indptr = torch.zeros(coo.n_rows + 1, dtype=torch.int64, device=coo.device)
    torch.cumsum(counts, dim=0, out=indptr[1:])
    return CSR(indptr, coo.cols, coo.values, coo.n_rows, coo.n_cols)


def csr_to_coo(csr: CSR) -> COO:
    """CSR -> COO expansion (reference convert)."""
    lengths = (csr.indptr[1:] - csr.indptr[:-1]).to(torch.int64)
    rows = torch.repeat_interleave(
        torch.arange(csr.n_rows, device=csr.device, dtype=torch.int64), lengths)
    return COO(rows, csr.indices.clone(), csr.values.clone(), csr.n_rows, csr.n_cols)


def csr_to_dense(csr: CSR) -> torch.Tensor:
    """CSR -> dense matrix (reference csr2dense)."""
    return csr.to_torch_sparse().to_dense()


def dense_to_csr(d: torch.Tensor) -> CSR:
    """Dense -> CSR of the nonzero entries (reference dense2csr)."""
    return CSR.from_dense(d)


def adj_to_csr(adj: torch.Tensor) -> CSR:
    """Boolean adjacency matrix -> CSR with unit values (adj_to_csr.cuh:28-124)."""
    idx = adj.to(torch.bool).nonzero(as_tuple=True)
    coo = COO(idx[0], idx[1], torch.ones(idx[0].numel(), dtype=torch.float32,
                                         device=adj.device),
              adj.shape[0], adj.shape[1])
    return sorted_coo_to_csr(coo)


def bitmap_to_csr(bitmap: Bitset, n_rows: int, n_cols: int,
                  values: torch.Tensor | None = None) -> CSR:
    """Row-major bitmap of n_rows*n_cols bits -> CSR (bitmap_to_csr.cuh:33,125)."""
    dense = bitmap.to_dense()[: n_rows * n_cols].reshape(n_rows, n_cols)
    idx = dense.nonzero(as_tuple=True)
    vals = (values if values is not None
            else torch.ones(idx[0].numel(), dtype=torch.float32, device=dense.device))
    return sorted_coo_to_csr(COO(idx[0], idx[1], vals, n_rows, n_cols))


def bitset_to_csr(bitset: Bitset, n_cols: int) -> CSR:
    """One-row CSR from a bitset (reference bitset_to_csr)."""
    return bitmap_to_csr(bitset, 1, n_cols)
