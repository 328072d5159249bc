"""Matrix-vector broadcast ops (linewise).

Reference parity: raft/linalg/matrix_vector_op.cuh (broadcast a vector along
rows or columns with a binary op) and matrix/linewise_op
(detail/linewise_op.cuh:252-446, the vectorized row/col broadcast engine).

MI355X design: single-vector broadcasts route to the native float4 kernel
(csrc/linewise.hip); TWO-stage chains like (x - mu) / sigma run fused in ONE
HBM pass via linewise_fused — two torch broadcasts are two full passes of a
pure-bandwidth op.
"""
from __future__ import annotations

from typing import Callable

import torch

_OPS = {
    "add": torch.add,
    "sub": torch.sub,
    "mul": torch.mul,
    "div": torch.div,
}
_OP_CODES = {"add": 0, "sub": 1, "mul": 2, "div": 3}


def _native_ok(mat, *vecs):
    return (mat.is_cuda and mat.dtype == torch.float32 and mat.dim() == 2
            and mat.is_contiguous()
            and all(v.dtype == torch.float32 for v in vecs))


def matrix_vector_op(mat: torch.Tensor, vec: torch.Tensor, op="add",
                     along_rows: bool = True, out: torch.Tensor | None = None) -> torch.Tensor:
    """Apply `op(mat_row_or_col, vec)` broadcast along rows (vec len = n_cols)
    or along columns (vec len = n_rows)."""
    assert mat.dim() == 2 and vec.dim() == 1
    if isinstance(op, str) and op in _OP_CODES and _native_ok(mat, vec):
        from raft_amd._ext import require_ext
        res = require_ext().linewise(mat, vec.contiguous(), None,
                                     bool(along_rows), _OP_CODES[op], 0)
        if out is not None:
            out.copy_(res)
            return out
        return res
    fn: Callable = _OPS[op] if isinstance(op, str) else op
    if along_rows:
        assert vec.numel() == mat.shape[1], "vector length must equal n_cols"
        res = fn(mat, vec.unsqueeze(0))
    else:
        assert vec.numel() == mat.shape[0], "vector length must equal n_rows"
        res = fn(mat, vec.unsqueeze(1))
    if out is not None:
        out.copy_(res)
        return out
    return res


def linewise_fused(mat: torch.Tensor, v1: torch.Tensor, op1: str,
                   v2: torch.Tensor, op2: str,
                   along_rows: bool = True) -> torch.Tensor:
    """Fused two-stage broadcast: (mat op1 v1) op2 v2 in one pass.

    The standardization chain (x - mu) / sigma is the canonical use; on GPU
    this is ONE read + ONE write of the matrix instead of two of each.
    """
    if _native_ok(mat, v1, v2):
        from raft_amd._ext import require_ext
        return require_ext().linewise(mat, v1.contiguous(), v2.contiguous(),
                                      bool(along_rows), _OP_CODES[op1],
                                      _OP_CODES[op2])
    r = matrix_vector_op(mat, v1, op1, along_rows)
    return matrix_vector_op(r, v2, op2, along_rows)


def linewise_op(mat: torch.Tensor, *vecs: torch.Tensor, fn: Callable,
                along_rows: bool = True) -> torch.Tensor:
    """General multi-vector linewise op (matrix/detail/linewise_op.cuh)."""
    if along_rows:
        vs = [v.unsqueeze(0) for v in vecs]
    else:
        vs = [v.unsqueeze(1) for v in vecs]
    return fn(mat, *vs)
