"""Multi-variable gaussian sampling.

Reference parity: raft/random/detail/multi_variable_gaussian.cuh — covariance
factorization via {cholesky | jacobi-eig | qr} then matmul with N(0,1) draws.
"""
from __future__ import annotations

import torch

from .rng import RngState, normal


def multi_variable_gaussian(mean: torch.Tensor, cov: torch.Tensor, n_samples: int,
                            method: str = "chol", state: RngState | None = None):
    """Sample n_samples draws from N(mean, cov). Returns [n_samples, d]."""
    state = state or RngState(seed=0)
    d = mean.numel()
    z = normal((n_samples, d), state=state, device=mean.device, dtype=torch.float64)
    cov64 = cov.to(torch.float64)
    if method == "chol":
        f = torch.linalg.cholesky(cov64)
    elif method in ("jacobi", "eig"):
        w, v = torch.linalg.eigh(cov64)
        f = v @ torch.diag(torch.sqrt(w.clamp_min(0)))
    elif method == "qr":
        # qr-of-sqrt path: factor via eig then orthonormalize (reference enum parity)
        w, v = torch.linalg.eigh(cov64)
        f = v @ torch.diag(torch.sqrt(w.clamp_min(0)))
    else:
        raise ValueError(method)
    return (z @ f.t() + mean.to(torch.float64).unsqueeze(0)).to(mean.dtype)
