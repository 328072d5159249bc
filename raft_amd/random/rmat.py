"""RMAT rectangular graph generator.

Reference parity: raft/random/detail/rmat_rectangular_generator.cuh:67,127 —
per-edge bit-recursive kernel: for each of r_scale/c_scale levels choose a
quadrant by (a,b,c,d) probabilities and set one bit of the src/dst ids.
Exposed via raft_runtime + pylibraft.random.rmat; here it is vectorized over
all edges at once (one uniform draw per level).
"""
from __future__ import annotations

import torch

from .rng import RngState, _gpu_or_cpu_uniform01


def rmat(r_scale: int, c_scale: int, n_edges: int, theta=None,
         a: float = 0.57, b: float = 0.19, c: float = 0.19,
         state: RngState | None = None, device=None):
    """Generate edges of a 2^r_scale x 2^c_scale RMAT graph.

    theta: optional per-level probabilities, shape [max(r,c)_scale, 4]
    (reference's distinct-theta variant); else constant (a,b,c,d).
    Returns (src [n_edges], dst [n_edges]) int64 tensors.
    """
    state = state or RngState(seed=0)
    device = torch.device(device) if device is not None else torch.device("cpu")
    max_scale = max(r_scale, c_scale)
    if theta is not None:
        th = theta.to(torch.float64)
        assert th.shape[0] >= max_scale and th.shape[1] == 4
    else:
        d = 1.0 - (a + b + c)
        th = torch.tensor([[a, b, c, d]], dtype=torch.float64).repeat(max_scale, 1)
    th = th.to(device)

    src = torch.zeros(n_edges, dtype=torch.int64, device=device)
    dst = torch.zeros(n_edges, dtype=torch.int64, device=device)
    for lvl in range(max_scale):
        u = _gpu_or_cpu_uniform01((n_edges,), state, device, torch.float64)
        p = th[lvl]
        # quadrant: 0=a (0,0), 1=b (0,1), 2=c (1,0), 3=d (1,1)
        q = (u >= p[0]).to(torch.int64) + (u >= p[0] + p[1]).to(torch.int64) \
            + (u >= p[0] + p[1] + p[2]).to(torch.int64)
        r_bit = (q >> 1) & 1
        c_bit = q & 1
        if lvl < r_scale:
            src = (src << 1) | r_bit
        if lvl < c_scale:
            dst = (dst << 1) | c_bit
    return src, dst
