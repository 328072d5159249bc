"""Sparse randomized SVD.

Reference parity: raft/sparse/solver/randomized_svds (range finder over SpMM +
cholesky_qr + small svd + sign correction svds_sign_correction.cuh), exposed
as pylibraft.sparse.linalg.svds.
"""
from __future__ import annotations

import torch

from raft_amd.random.rng import RngState, normal
from raft_amd.linalg.decomp import svd_flip
from ..types import CSR, COO
from ..convert import coo_to_csr
from ..linalg import spmm, csr_transpose


def _cholesky_qr(y: torch.Tensor) -> torch.Tensor:
    """Q factor via Cholesky QR (reference cholesky_qr.cuh): Y^T Y = R^T R,
    Q = Y R^-1 — one gemm + small cholesky, GPU-friendly."""
    g = y.t() @ y
    # jitter for numerical safety
    g = g + torch.eye(g.shape[0], device=y.device, dtype=y.dtype) * (
        torch.diagonal(g).max() * 1e-10)
    r = torch.linalg.cholesky(g, upper=True)
    return y @ torch.linalg.inv(r)


def randomized_svds(a, k: int, p: int = 10, n_iter: int = 4, seed: int = 42):
    """Approximate top-k SVD of sparse A [m, n]. Returns (U, S, V)."""
    if isinstance(a, COO):
        a = coo_to_csr(a)
    assert isinstance(a, CSR)
    m, n = a.n_rows, a.n_cols
    l = min(k + p, min(m, n))
    state = RngState(seed=seed)
    at = csr_transpose(a)
    omega = normal((n, l), state=state, device=a.device, dtype=a.values.dtype)
    y = spmm(a, omega)
    q = _cholesky_qr(y)
    for _ in range(n_iter):
        z = spmm(at, q)
        qz = _cholesky_qr(z)
        y = spmm(a, qz)
        q = _cholesky_qr(y)
    b = spmm(at, q).t()          # l x n
    ub, s, vh = torch.linalg.svd(b, full_matrices=False)
    u = q @ ub
    u, v = svd_flip(u[:, :k], vh.t()[:, :k])
    return u, s[:k], v
