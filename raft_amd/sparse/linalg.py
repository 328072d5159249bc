"""Sparse linear algebra: SpMV/SpMM/SDDMM, Laplacian, symmetrize, norms.

Reference parity: raft/sparse/linalg/* (cusparse SpMV/SpMM/SDDMM wrappers,
laplacian.cuh, symmetrize.cuh, norm/degree/add, masked_matmul).

MI355X: the hot CSR SpMV (Lanczos inner loop, BASELINE config 4) is a
hand-written HIP kernel (csrc/spmv.hip) — one wave per row for long rows /
row-per-lane for short rows, vectorized value loads, HBM-bandwidth-bound by
design (~12 bytes/nnz). Other ops compose torch.sparse (rocSPARSE).
"""
from __future__ import annotations

import torch

from raft_amd._ext import require_ext
from raft_amd.core.bitset import Bitset
from .types import CSR, COO
from .convert import csr_to_coo, coo_to_csr, sorted_coo_to_csr


def spmv(a, x: torch.Tensor, out: torch.Tensor | None = None) -> torch.Tensor:
    """y = A @ x for CSR or COO A [m,n], dense x [n] (reference SpMV accepts
    both container types)."""
    if isinstance(a, COO):
        a = coo_to_csr(a)
    assert x.dim() == 1 and x.numel() == a.n_cols
    if a.values.is_cuda and a.values.dtype in (torch.float32, torch.float64):
        ext = require_ext()
        y = ext.csr_spmv(a.indptr.to(torch.int32).contiguous(),
                         a.indices.to(torch.int32).contiguous(),
                         a.values.contiguous(), x.contiguous(), int(a.n_rows))
        if out is not None:
            out.copy_(y)
            return out
        return y
    y = a.to_torch_sparse() @ x.unsqueeze(1)
    y = y.squeeze(1)
    if out is not None:
        out.copy_(y)
        return out
    return y


def spmm(a, b: torch.Tensor) -> torch.Tensor:
    """C = A @ B for dense B [n, k] (rocSPARSE via torch.sparse); CSR or COO."""
    if isinstance(a, COO):
        a = coo_to_csr(a)
    return a.to_torch_sparse() @ b


def sddmm(a: torch.Tensor, b: torch.Tensor, mask: CSR) -> CSR:
    """Sampled dense-dense matmul: (A @ B^T) restricted to mask's pattern.

    GPU: native sub-wave-per-edge kernel (csrc/spmv.hip) — no [nnz, d]
    gather temporaries (the CPU composition materializes 2*nnz*d floats)."""
    coo = csr_to_coo(mask)
    if a.is_cuda and a.dtype == torch.float32 and b.dtype == torch.float32:
        from raft_amd._ext import require_ext
        vals = require_ext().sddmm(a.contiguous(), b.contiguous(),
                                   coo.rows.to(torch.int32),
                                   coo.cols.to(torch.int32))
        return CSR(mask.indptr, mask.indices, vals, mask.n_rows, mask.n_cols)
    vals = (a[coo.rows.to(torch.int64)] * b[coo.cols.to(torch.int64)]).sum(dim=1)
    return CSR(mask.indptr, mask.indices, vals, mask.n_rows, mask.n_cols)


def masked_matmul(a: torch.Tensor, b: torch.Tensor, bitmask: Bitset) -> CSR:
    """A @ B^T under a bitmap sparsity mask -> CSR (masked_matmul.cuh:47-92:
    bitmap->CSR + SDDMM composition)."""
    from .convert import bitmap_to_csr
    mask = bitmap_to_csr(bitmask, a.shape[0], b.shape[0])
    return sddmm(a, b, mask)


def csr_degree(a: CSR) -> torch.Tensor:
    """Per-row nonzero counts (reference degree kernels)."""
    return (a.indptr[1:] - a.indptr[:-1]).to(torch.int64)


def csr_row_norm(a: CSR, norm_type: str = "l2") -> torch.Tensor:
    """Per-row L1/L2/Linf norms of CSR values (reference sparse norm)."""
    seg = torch.repeat_interleave(torch.arange(a.n_rows, device=a.device),
                                  (a.indptr[1:] - a.indptr[:-1]).to(torch.int64))
    out = torch.zeros(a.n_rows, dtype=a.values.dtype, device=a.device)
    if norm_type == "l2":
        out.index_add_(0, seg, a.values * a.values)
    elif norm_type == "l1":
        out.index_add_(0, seg, a.values.abs())
    elif norm_type == "linf":
        out.scatter_reduce_(0, seg, a.values.abs(), reduce="amax")
    else:
        raise ValueError(norm_type)
    return out


def csr_add(a: CSR, b: CSR) -> CSR:
    """CSR + CSR (2-pass nnz then fill in the reference; torch handles dedup)."""
    assert (a.n_rows, a.n_cols) == (b.n_rows, b.n_cols)
    t = (a.to_torch_sparse().to_sparse_coo() + b.to_torch_sparse().to_sparse_coo()).coalesce()
    coo = COO(t.indices()[0], t.indices()[1], t.values(), a.n_rows, a.n_cols)
    return sorted_coo_to_csr(coo)


def csr_transpose(a: CSR) -> CSR:
    """CSR transpose (rocSPARSE csr2csc analog)."""
    coo = csr_to_coo(a)
    return coo_to_csr(COO(coo.cols, coo.rows, coo.values, a.n_cols, a.n_rows))


def symmetrize_coo(coo: COO, op: str = "add") -> COO:
    """A (+|max|min) A^T with duplicate coalescing (symmetrize.cuh:29-53)."""
    rows = torch.cat([coo.rows, coo.cols])
    cols = torch.cat([coo.cols, coo.rows])
    vals = torch.cat([coo.values, coo.values])
    key = rows.to(torch.int64) * coo.n_cols + cols.to(torch.int64)
    uniq, inv = torch.unique(key, return_inverse=True)
    out_v = torch.zeros(uniq.numel(), dtype=vals.dtype, device=vals.device)
    if op == "add":
        out_v.index_add_(0, inv, vals)
    elif op == "max":
        out_v.fill_(float("-inf"))
        out_v.scatter_reduce_(0, inv, vals, reduce="amax")
    elif op == "mean":
        out_v.index_add_(0, inv, vals)
        cnt = torch.zeros_like(out_v).index_add_(0, inv, torch.ones_like(vals))
        out_v = out_v / cnt
    else:
        raise ValueError(op)
    return COO(uniq // coo.n_cols, uniq % coo.n_cols, out_v,
               max(coo.n_rows, coo.n_cols), max(coo.n_rows, coo.n_cols))


def knn_graph_symmetrize(knn_idx: torch.Tensor, knn_dist: torch.Tensor) -> COO:
    """Symmetrize a knn graph given [n,k] neighbor idx/dist (symmetrize.cuh:57-137)."""
    n, k = knn_idx.shape
    rows = torch.repeat_interleave(torch.arange(n, device=knn_idx.device), k)
    coo = COO(rows, knn_idx.reshape(-1).to(torch.int64), knn_dist.reshape(-1), n, n)
    return symmetrize_coo(coo, op="max")


def laplacian(a: CSR) -> CSR:
    """L = D - A (compute_graph_laplacian_kernel, laplacian.cuh:41)."""
    deg = _weighted_degree(a)
    coo = csr_to_coo(a)
    rows = torch.cat([coo.rows, torch.arange(a.n_rows, device=a.device)])
    cols = torch.cat([coo.cols, torch.arange(a.n_rows, device=a.device)])
    vals = torch.cat([-coo.values, deg])
    return coo_to_csr(COO(rows, cols, vals, a.n_rows, a.n_cols))


def laplacian_normalized(a: CSR) -> CSR:
    """L_sym = I - D^-1/2 A D^-1/2 (laplacian.cuh:237)."""
    deg = _weighted_degree(a)
    dinv = torch.where(deg > 0, deg.pow(-0.5), torch.zeros_like(deg))
    coo = csr_to_coo(a)
    vals = -coo.values * dinv[coo.rows.to(torch.int64)] * dinv[coo.cols.to(torch.int64)]
    eye_r = torch.arange(a.n_rows, device=a.device)
    ones = (deg > 0).to(a.values.dtype)
    rows = torch.cat([coo.rows, eye_r])
    cols = torch.cat([coo.cols, eye_r])
    v = torch.cat([vals, ones])
    return coo_to_csr(COO(rows, cols, v, a.n_rows, a.n_cols))


def _weighted_degree(a: CSR) -> torch.Tensor:
    seg = torch.repeat_interleave(torch.arange(a.n_rows, device=a.device),
                                  (a.indptr[1:] - a.indptr[:-1]).to(torch.int64))
    deg = torch.zeros(a.n_rows, dtype=a.values.dtype, device=a.device)
    deg.index_add_(0, seg, a.values)
    return deg
