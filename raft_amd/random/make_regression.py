"""Linear-model synthetic data (reference: raft/random/make_regression.cuh —
gemm of random matrices, optional effective-rank low-rank profile, noise,
shuffle, optional bias)."""
from __future__ import annotations

import torch

from .rng import RngState, normal, uniform
from .permute import permute


def _make_low_rank_matrix(n, d, effective_rank, tail_strength, state, device):
    """Low-rank matrix with bell-shaped singular profile
    (detail/make_regression.cuh:34-47)."""
    k = min(n, d)
    u, _ = torch.linalg.qr(normal((n, k), state=state, device=device, dtype=torch.float64))
    v, _ = torch.linalg.qr(normal((d, k), state=state, device=device, dtype=torch.float64))
    idx = torch.arange(k, device=device, dtype=torch.float64) / max(effective_rank, 1)
    low = (1 - tail_strength) * torch.exp(-(idx ** 2))
    tail = tail_strength * torch.exp(-0.1 * idx)
    s = low + tail
    return (u * s.unsqueeze(0)) @ v.t()


def make_regression(n_rows: int, n_cols: int, n_informative: int | None = None,
                    n_targets: int = 1, bias: float = 0.0, noise: float = 0.0,
                    effective_rank: int | None = None, tail_strength: float = 0.5,
                    shuffle: bool = False, state: RngState | None = None,
                    device=None, dtype=torch.float32):
    """Returns (X, y, coef). y = X @ coef + bias + N(0, noise)."""
    state = state or RngState(seed=0)
    device = torch.device(device) if device is not None else torch.device("cpu")
    n_informative = n_informative if n_informative is not None else n_cols

    if effective_rank is None:
        x = normal((n_rows, n_cols), state=state, device=device, dtype=torch.float64)
    else:
        x = _make_low_rank_matrix(n_rows, n_cols, effective_rank, tail_strength, state, device)

    coef = torch.zeros((n_cols, n_targets), device=device, dtype=torch.float64)
    coef[:n_informative] = 100.0 * uniform((n_informative, n_targets), state=state,
                                           device=device, dtype=torch.float64)
    y = x @ coef + bias
    if noise > 0:
        y = y + normal(tuple(y.shape), 0.0, noise, state=state, device=device, dtype=torch.float64)
    if shuffle:
        perm = permute(n_rows, state=state, device=device)
        x, y = x[perm], y[perm]
    y = y.squeeze(1) if n_targets == 1 else y
    return x.to(dtype), y.to(dtype), coef.to(dtype)
