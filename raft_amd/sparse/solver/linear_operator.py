"""Abstract SpMV operator for iterative solvers
(reference: raft/sparse/solver/csr_linear_operator)."""
from __future__ import annotations

from typing import Callable

import torch

from ..types import CSR
from ..linalg import spmv


class LinearOperator:
    def __init__(self, shape: tuple, matvec: Callable[[torch.Tensor], torch.Tensor],
                 device=None, dtype=torch.float32):
        self.shape = shape
        self._matvec = matvec
        self.device = device
        self.dtype = dtype

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        return self._matvec(x)


def csr_operator(a: CSR) -> LinearOperator:
    return LinearOperator((a.n_rows, a.n_cols), lambda x: spmv(a, x),
                          device=a.device, dtype=a.values.dtype)
