"""Row/column norms and fused row-normalize.

Reference parity: raft/linalg/norm.cuh (L0/L1/L2/Linf via reduce+final-op,
norm_types.hpp:40-49) and detail/normalize.cuh (fused row-normalize, thin
logical-warp + block-per-row kernels).

MI355X: row norms run the native wave64 row-reduction kernel (sq/abs main ops
+ sqrt final op fused at the epilogue); normalize is a single fused kernel
(norm + scale in one HBM pass) in csrc/reductions.hip.
"""
from __future__ import annotations

from enum import Enum

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu
from .reduce import coalesced_reduction, strided_reduction


class NormType(Enum):
    L0 = "l0"          # count of non-zeros
    L1 = "l1"
    L2 = "l2"          # sqrt of sum of squares (when sqrt=True)
    LINF = "linf"


def row_norm(x: torch.Tensor, norm_type: NormType = NormType.L2, sqrt: bool = True) -> torch.Tensor:
    """Per-row norms (native fused kernel on GPU)."""
    return _norm_along(x, norm_type, sqrt, along_rows=True)


def col_norm(x: torch.Tensor, norm_type: NormType = NormType.L2, sqrt: bool = True) -> torch.Tensor:
    """Per-column norms (reduce + final op)."""
    return _norm_along(x, norm_type, sqrt, along_rows=False)


def _norm_along(x, norm_type, sqrt, along_rows: bool):
    red = coalesced_reduction if along_rows else strided_reduction
    if norm_type == NormType.L2:
        return red(x, main_op="sq", reduce_op="sum", final_op="sqrt" if sqrt else "identity")
    if norm_type == NormType.L1:
        return red(x, main_op="abs", reduce_op="sum")
    if norm_type == NormType.LINF:
        return red(x, main_op="abs", reduce_op="max")
    if norm_type == NormType.L0:
        dim = 1 if along_rows else 0
        return (x != 0).sum(dim=dim).to(x.dtype)
    raise ValueError(norm_type)


def norm(x: torch.Tensor, norm_type: NormType = NormType.L2, along_rows: bool = True,
         sqrt: bool = True) -> torch.Tensor:
    """Row or column norms with L0/L1/L2/Linf types (reference norm.cuh)."""
    return _norm_along(x, norm_type, sqrt, along_rows)


def normalize(x: torch.Tensor, norm_type: NormType = NormType.L2, eps: float = 1e-12,
              out: torch.Tensor | None = None) -> torch.Tensor:
    """Row-normalize (detail/normalize.cuh). Fused single-pass kernel on GPU."""
    assert x.dim() == 2
    if on_gpu(x) and norm_type == NormType.L2 and x.dtype == torch.float32:
        ext = require_ext()
        return ext.row_normalize_l2(x.contiguous(), float(eps))
    n = _norm_along(x, norm_type, sqrt=True, along_rows=True)
    n = torch.clamp(n, min=eps)
    res = x / n.unsqueeze(1)
    if out is not None:
        out.copy_(res)
        return out
    return res
