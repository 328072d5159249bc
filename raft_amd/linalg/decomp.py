"""Dense factorizations: eig/svd/qr/cholesky/lstsq.

Reference parity: raft/linalg/{eig,svd,qr,lstsq,cholesky_r1_update}.cuh —
cuSOLVER wrappers (syevd/syevdx/syevj, gesvd/gesvdj, geqrf+orgqr, 4 lstsq
solvers, rank-1 Cholesky update).

MI355X: torch.linalg on ROCm is the rocSOLVER/hipSOLVER vendor path — the
direct analog of the reference's cuSOLVER usage. The 4 lstsq variants and the
rank-1 Cholesky update are composed here exactly as the reference composes
them from the vendor primitives.
"""
from __future__ import annotations

import torch


def eigh(a: torch.Tensor, uplo: str = "L"):
    """Symmetric eigendecomposition (detail/eig.cuh:39-76 syevd analog).
    Returns (eigenvalues ascending, eigenvectors)."""
    w, v = torch.linalg.eigh(a, UPLO=uplo)
    return w, v


#: reference spells it `eig_dc`; keep a simple alias
eig = eigh


def eig_jacobi(a: torch.Tensor, tol: float = 1e-7, max_sweeps: int = 15):
    """Jacobi eigensolver (syevj analog). rocSOLVER's syevj is reached through
    the same torch.linalg.eigh entry; tol/max_sweeps kept for API parity."""
    return torch.linalg.eigh(a)


def eig_selective(a: torch.Tensor, n_eig: int, largest: bool = True):
    """Selective eigendecomposition (syevdx analog): top/bottom n_eig pairs."""
    w, v = torch.linalg.eigh(a)
    if largest:
        return w[-n_eig:], v[:, -n_eig:]
    return w[:n_eig], v[:, :n_eig]


def svd(a: torch.Tensor, full_matrices: bool = False):
    """SVD (detail/svd.cuh gesvd analog). Returns (U, S, V) with A = U S V^T."""
    u, s, vh = torch.linalg.svd(a, full_matrices=full_matrices)
    return u, s, vh.t().conj() if a.is_complex() else vh.t()


def svd_flip(u: torch.Tensor, v: torch.Tensor):
    """Deterministic sign correction (reference svd_flip): make the max-|.|
    element of each U column positive."""
    idx = u.abs().argmax(dim=0)
    signs = torch.sign(u[idx, torch.arange(u.shape[1], device=u.device)])
    signs = torch.where(signs == 0, torch.ones_like(signs), signs)
    return u * signs.unsqueeze(0), v * signs.unsqueeze(0)


def qr(a: torch.Tensor, mode: str = "reduced"):
    """QR (detail/qr.cuh geqrf+orgqr analog)."""
    return torch.linalg.qr(a, mode=mode)


def cholesky(a: torch.Tensor, upper: bool = False) -> torch.Tensor:
    """Cholesky factor (rocSOLVER potrf analog via torch.linalg)."""
    l = torch.linalg.cholesky(a)
    return l.t() if upper else l


def cholesky_r1_update(l: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """Rank-1 update of a Cholesky factor: chol(A + x x^T) from L=chol(A).

    Reference parity: raft/linalg/cholesky_r1_update.cuh. Classic hyperbolic
    rotation scheme, O(n^2). On GPU this is ONE kernel launch
    (csrc/solver_kernels.hip) — the per-k Python loop cost n x ~6 launch
    round-trips.
    """
    l = l.clone()
    x = x.clone().to(l.dtype)
    n = l.shape[0]
    if l.is_cuda and l.dtype in (torch.float32, torch.float64) and n > 1:
        from raft_amd._ext import require_ext
        lc = l.contiguous()
        require_ext().cholesky_r1_update_(lc, x.contiguous())
        return lc
    for k in range(n):
        lkk = l[k, k]
        r = torch.sqrt(lkk * lkk + x[k] * x[k])
        c = r / lkk
        s = x[k] / lkk
        l[k, k] = r
        if k + 1 < n:
            l[k + 1:, k] = (l[k + 1:, k] + s * x[k + 1:]) / c
            x[k + 1:] = c * x[k + 1:] - s * l[k + 1:, k]
    return l


def lstsq(a: torch.Tensor, b: torch.Tensor, algo: str = "qr") -> torch.Tensor:
    """Least squares min ||A w - b||.

    Reference parity: detail/lstsq.cuh's four solvers — lstsqSvdQR (:111),
    lstsqSvdJacobi (:171), lstsqEig (Gram-matrix eig, :242), lstsqQR (:346).
    """
    if algo in ("svd-qr", "svd-jacobi"):
        u, s, vt = torch.linalg.svd(a, full_matrices=False)
        s_inv = torch.where(s > s.max() * 1e-7, 1.0 / s, torch.zeros_like(s))
        return vt.t() @ (s_inv * (u.t() @ b))
    if algo == "eig":
        # Gram matrix path: (A^T A) w = A^T b via eigendecomposition
        g = a.t() @ a
        w, v = torch.linalg.eigh(g)
        w_inv = torch.where(w > w.max() * 1e-7, 1.0 / w, torch.zeros_like(w))
        return v @ (w_inv * (v.t() @ (a.t() @ b)))
    if algo == "qr":
        return torch.linalg.lstsq(a, b).solution
    raise ValueError(f"unknown lstsq algo {algo}")
