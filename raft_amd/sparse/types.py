"""Sparse containers (reference: raft/core/sparse_types.hpp:91,
device_csr_matrix.hpp, device_coo_matrix.hpp, sparse/coo.hpp).

Thin dataclasses over torch tensors: indptr/indices int32 (rocSPARSE-native
index width; int64 accepted), values any float dtype, device = wherever the
tensors live.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class CSR:
    indptr: torch.Tensor    # [n_rows + 1]
    indices: torch.Tensor   # [nnz]
    values: torch.Tensor    # [nnz]
    n_rows: int
    n_cols: int

    @property
    def nnz(self) -> int:
        return int(self.values.numel())

    @property
    def device(self):
        return self.values.device

    @property
    def dtype(self):
        return self.values.dtype

    def to(self, device) -> "CSR":
        return CSR(self.indptr.to(device), self.indices.to(device),
                   self.values.to(device), self.n_rows, self.n_cols)

    def to_torch_sparse(self) -> torch.Tensor:
        return torch.sparse_csr_tensor(self.indptr.to(torch.int64),
                                       self.indices.to(torch.int64),
                                       self.values, size=(self.n_rows, self.n_cols))

    @classmethod
    def from_torch_sparse(cls, t: torch.Tensor) -> "CSR":
        t = t.to_sparse_csr() if t.layout != torch.sparse_csr else t
        return cls(t.crow_indices(), t.col_indices(), t.values(),
                   t.shape[0], t.shape[1])

    @classmethod
    def from_dense(cls, d: torch.Tensor) -> "CSR":
        return cls.from_torch_sparse(d.to_sparse_csr())

    def row_lengths(self) -> torch.Tensor:
        return self.indptr[1:] - self.indptr[:-1]


@dataclass
class COO:
    rows: torch.Tensor
    cols: torch.Tensor
    values: torch.Tensor
    n_rows: int
    n_cols: int

    @property
    def nnz(self) -> int:
        return int(self.values.numel())

    @property
    def device(self):
        return self.values.device

    def to(self, device) -> "COO":
        return COO(self.rows.to(device), self.cols.to(device),
                   self.values.to(device), self.n_rows, self.n_cols)

    def to_dense(self) -> torch.Tensor:
        out = torch.zeros((self.n_rows, self.n_cols), dtype=self.values.dtype,
                          device=self.device)
        out.index_put_((self.rows.to(torch.int64), self.cols.to(torch.int64)),
                       self.values, accumulate=True)
        return out
