"""Randomized SVD (range finder + power iterations + small SVD).

Reference parity: raft/linalg/rsvd.cuh:41-455 / detail/rsvd.cuh — random
projection range finder, optional power iterations with QR stabilization,
then exact SVD of the projected matrix; fixed-rank and fixed-percent variants.
"""
from __future__ import annotations

import torch

from .decomp import qr, svd_flip


def rsvd(a: torch.Tensor, k: int, p: int = 10, n_iter: int = 2,
         seed: int | None = None):
    """Approximate rank-k SVD of a (m x n). Returns (U, S, V)."""
    m, n = a.shape
    l = min(k + p, min(m, n))
    gen = None
    if seed is not None:
        gen = torch.Generator(device="cpu").manual_seed(seed)
    omega = torch.randn(n, l, generator=gen, dtype=a.dtype).to(a.device)
    y = a @ omega
    q, _ = qr(y)
    for _ in range(n_iter):
        z = a.t() @ q
        qz, _ = qr(z)
        y = a @ qz
        q, _ = qr(y)
    b = q.t() @ a           # l x n
    ub, s, vh = torch.linalg.svd(b, full_matrices=False)
    u = q @ ub
    u, v = svd_flip(u[:, :k], vh.t()[:, :k])
    return u, s[:k], v


def rsvd_fixed_rank(a, k, p=10, n_iter=2, seed=None):
    return rsvd(a, k, p=p, n_iter=n_iter, seed=seed)


def rsvd_percent(a, percent: float, p: int = 10, n_iter: int = 2, seed=None):
    k = max(1, int(min(a.shape) * percent))
    return rsvd(a, k, p=p, n_iter=n_iter, seed=seed)
