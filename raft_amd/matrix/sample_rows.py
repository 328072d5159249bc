"""Uniform row subsampling (reference: raft/matrix/sample_rows.cuh —
rng excess-sampling + gather)."""
from __future__ import annotations

import torch

from raft_amd.random.rng import RngState, sample_without_replacement
from .gather import gather


def sample_rows(x: torch.Tensor, n_samples: int, state: RngState | None = None) -> torch.Tensor:
    """Uniform row subsample without replacement (reference sample_rows)."""
    if state is None:
        state = RngState(seed=0)
    idx = sample_without_replacement(x.shape[0], n_samples, state=state, device=x.device)
    return gather(x, idx)
