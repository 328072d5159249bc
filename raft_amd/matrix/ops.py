"""Structural + math matrix ops.

Reference parity: raft/matrix/detail/matrix.cuh (copyRows/slice/triangular/
diagonal/reverse), detail/shift.cuh, detail/math.cuh (power/ratio/reciprocal/
sqrt/sign_flip/threshold), linewise_op.
"""
from __future__ import annotations

import torch


def slice_matrix(x: torch.Tensor, r0: int, c0: int, r1: int, c1: int) -> torch.Tensor:
    """Rectangular sub-matrix copy (reference slice)."""
    return x[r0:r1, c0:c1].contiguous()


def get_diagonal(x: torch.Tensor) -> torch.Tensor:
    """Copy of the main diagonal (reference diagonal copy)."""
    return torch.diagonal(x).contiguous()


def set_diagonal(x: torch.Tensor, vec: torch.Tensor) -> torch.Tensor:
    """Write vec onto the main diagonal in place."""
    n = min(x.shape)
    idx = torch.arange(n, device=x.device)
    x[idx, idx] = vec[:n].to(x.dtype)
    return x


def upper_triangular(x: torch.Tensor) -> torch.Tensor:
    """Upper-triangular copy (reference upper-triangular copy)."""
    return torch.triu(x)


def lower_triangular(x: torch.Tensor) -> torch.Tensor:
    """Lower-triangular copy (reference tril analog)."""
    return torch.tril(x)


def row_reverse(x: torch.Tensor) -> torch.Tensor:
    """Reverse row order (reference rowReverse)."""
    return torch.flip(x, dims=[0])


def col_reverse(x: torch.Tensor) -> torch.Tensor:
    """Reverse column order (reference colReverse)."""
    return torch.flip(x, dims=[1])


def shift_rows(x: torch.Tensor, k: int, fill_value: float = 0.0) -> torch.Tensor:
    """Shift rows down by k (k<0: up), filling vacated rows (detail/shift.cuh)."""
    out = torch.full_like(x, fill_value)
    n = x.shape[0]
    if k >= 0:
        out[k:] = x[: n - k]
    else:
        out[:k] = x[-k:]
    return out


def eye(n: int, m: int | None = None, device=None, dtype=torch.float32) -> torch.Tensor:
    """Identity matrix (reference eye initializer)."""
    return torch.eye(n, m if m is not None else n, device=device, dtype=dtype)


def power(x: torch.Tensor, p: float) -> torch.Tensor:
    """Elementwise power (reference matrix power op)."""
    return torch.pow(x, p)


def ratio(x: torch.Tensor) -> torch.Tensor:
    """Normalize entries to sum to 1 (reference matrix::ratio)."""
    return x / x.sum()


def reciprocal(x: torch.Tensor, scalar: float = 1.0, thres: float = 0.0) -> torch.Tensor:
    """scalar / x where |x| > thres else 0 (guarded reciprocal)."""
    out = torch.where(x.abs() > thres, scalar / x, torch.zeros_like(x))
    return out


def sqrt(x: torch.Tensor) -> torch.Tensor:
    """Elementwise sqrt (exported as matrix_sqrt; reference matrix sqrt op)."""
    return torch.sqrt(x)


def sign_flip(x: torch.Tensor) -> torch.Tensor:
    """Flip column signs so each column's max-|.| element is positive."""
    idx = x.abs().argmax(dim=0)
    signs = torch.sign(x[idx, torch.arange(x.shape[1], device=x.device)])
    signs = torch.where(signs == 0, torch.ones_like(signs), signs)
    return x * signs.unsqueeze(0)


def threshold(x: torch.Tensor, thres: float) -> torch.Tensor:
    """Zero out entries below thres (reference threshold op)."""
    return torch.where(x < thres, torch.zeros_like(x), x)


def linewise(x: torch.Tensor, vec: torch.Tensor, fn, along_rows: bool = True) -> torch.Tensor:
    """Broadcast a vector op along rows/columns (reference linewise_op)."""
    from raft_amd.linalg.matrix_vector import matrix_vector_op
    return matrix_vector_op(x, vec, op=fn, along_rows=along_rows)
