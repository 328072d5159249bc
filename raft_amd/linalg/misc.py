"""Small linalg utilities: MSE, initializers, transpose.

Reference parity: raft/linalg/mean_squared_error.cuh, init.cuh (iota/eye),
detail/transpose.cuh (geam for fp / tiled-smem kernel for non-fp — on ROCm
torch's .t().contiguous() emits the vendor transpose kernel).
"""
from __future__ import annotations

import torch


def mean_squared_error(a: torch.Tensor, b: torch.Tensor, weight: float = 1.0) -> torch.Tensor:
    """Weighted mean squared error (reference mean_squared_error.cuh)."""
    return ((a - b) ** 2).mean() * weight


def init_iota(n: int, start: float = 0.0, step: float = 1.0, device=None,
              dtype=torch.float32) -> torch.Tensor:
    """Arithmetic-sequence initializer (reference init.cuh iota)."""
    return torch.arange(n, device=device, dtype=dtype) * step + start


def init_eye(n: int, m: int | None = None, device=None, dtype=torch.float32) -> torch.Tensor:
    """Identity matrix initializer (reference init.cuh eye)."""
    return torch.eye(n, m if m is not None else n, device=device, dtype=dtype)


def transpose(a: torch.Tensor) -> torch.Tensor:
    """Matrix transpose, contiguous result (reference linalg::transpose)."""
    return a.t().contiguous()
