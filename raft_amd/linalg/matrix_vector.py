"""Matrix-vector broadcast ops (linewise).

Reference parity: raft/linalg/matrix_vector_op.cuh (broadcast a vector along
rows or columns with a binary op) and matrix/linewise_op (detail/linewise_op.cuh,
the vectorized row/col broadcast engine).

Torch broadcasting emits a single vectorized HIP kernel for these shapes, which
is exactly the linewise engine's job — no bespoke kernel needed (memory-bound).
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


def matrix_vector_op(mat: torch.Tensor, vec: torch.Tensor, op="add",
                     along_rows: bool = True, out: torch.Tensor | None = None) -> torch.Tensor:
    """Apply `op(mat_row_or_col, vec)` broadcast along rows (vec len = n_cols)
    or along columns (vec len = n_rows)."""
    fn: Callable = _OPS[op] if isinstance(op, str) else op
    assert mat.dim() == 2 and vec.dim() == 1
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


def linewise_op(mat: torch.Tensor, *vecs: torch.Tensor, fn: Callable,
                along_rows: bool = True) -> torch.Tensor:
    """General multi-vector linewise op (matrix/detail/linewise_op.cuh)."""
    if along_rows:
        vs = [v.unsqueeze(0) for v in vecs]
    else:
        vs = [v.unsqueeze(1) for v in vecs]
    return fn(mat, *vs)
