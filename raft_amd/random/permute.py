"""Random permutation (reference: raft/random/permute.cuh — coalesced
LCG-derived permutation kernel). Here: key-sort of counter-RNG draws, which is
bias-free and fully parallel (rocPRIM radix sort under torch.argsort)."""
from __future__ import annotations

import torch

from .rng import RngState, _draw_u32, require_ext


def permute(n: int, state: RngState | None = None, device=None) -> torch.Tensor:
    """Random permutation of 0..n-1 (counter-based RNG)."""
    state = state or RngState(seed=0)
    device = torch.device(device) if device is not None else torch.device("cpu")
    if device.type == "cuda":
        ext = require_ext()
        keys = ext.rng_uniform(n, int(state.seed), int(state.base_subsequence), device.index or 0)
        state.advance(1)
        return torch.argsort(keys)
    keys = _draw_u32(n, state, device, n_draws=1)[0]
    return torch.argsort(keys)


def permute_rows(x: torch.Tensor, state: RngState | None = None) -> torch.Tensor:
    """Randomly permute the rows of x (reference permute)."""
    perm = permute(x.shape[0], state=state, device=x.device)
    return x[perm]
