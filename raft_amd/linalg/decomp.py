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


def eig_jacobi(a: torch.Tensor, tol: float = 1e-7, max_sweeps: int = 30):
    """Jacobi eigensolver (detail/eig.cuh:276-293 syevj analog) — a REAL
    cyclic-Jacobi implementation, not an eigh alias.

    Parallel one-sided ordering: each sweep processes n-1 rounds of a
    round-robin tournament pairing; every round rotates n/2 DISJOINT pivot
    pairs simultaneously (all rotations are independent, so the round is a
    batched tensor update — the same parallel-Jacobi structure syevj uses).
    Converges quadratically for symmetric matrices; stops when the
    off-diagonal Frobenius mass is below tol * ||A||_F or after max_sweeps.
    Returns (eigenvalues ascending, eigenvectors).
    """
    n = a.shape[0]
    assert a.shape == (n, n)
    work = a.to(torch.float64 if a.dtype == torch.float64 else torch.float32).clone()
    v = torch.eye(n, dtype=work.dtype, device=a.device)
    if n == 1:
        return work.diagonal().clone(), v
    # round-robin pairings: fix 0, rotate 1..n-1 (pad to even with a ghost)
    m = n + (n & 1)
    ring = list(range(1, n)) + ([n] if (n & 1) else [])  # n = ghost index
    norm_a = float(work.norm())
    for _ in range(max_sweeps):
        off = work.clone()
        off.diagonal().zero_()
        if float(off.norm()) <= tol * max(norm_a, 1e-300):
            break
        for _round in range(m - 1):
            seq = [0] + ring
            pairs = [(seq[i], seq[m - 1 - i]) for i in range(m // 2)]
            pairs = [(min(p, q), max(p, q)) for p, q in pairs if p < n and q < n]
            p_idx = torch.tensor([p for p, _ in pairs], device=a.device)
            q_idx = torch.tensor([q for _, q in pairs], device=a.device)
            app = work[p_idx, p_idx]
            aqq = work[q_idx, q_idx]
            apq = work[p_idx, q_idx]
            # rotation angles (vectorized over all disjoint pairs)
            tau = (aqq - app) / (2.0 * torch.where(apq == 0, torch.ones_like(apq), apq))
            t = torch.sign(tau) / (tau.abs() + torch.sqrt(1.0 + tau * tau))
            t = torch.where(apq == 0, torch.zeros_like(t), t)
            c = 1.0 / torch.sqrt(1.0 + t * t)
            s = t * c
            # apply J^T A J on rows/cols p,q and V J on cols p,q (disjoint
            # pairs -> one batched update per round)
            rp = work[p_idx, :].clone()
            rq = work[q_idx, :].clone()
            work[p_idx, :] = c.unsqueeze(1) * rp - s.unsqueeze(1) * rq
            work[q_idx, :] = s.unsqueeze(1) * rp + c.unsqueeze(1) * rq
            cp = work[:, p_idx].clone()
            cq = work[:, q_idx].clone()
            work[:, p_idx] = c.unsqueeze(0) * cp - s.unsqueeze(0) * cq
            work[:, q_idx] = s.unsqueeze(0) * cp + c.unsqueeze(0) * cq
            vp = v[:, p_idx].clone()
            vq = v[:, q_idx].clone()
            v[:, p_idx] = c.unsqueeze(0) * vp - s.unsqueeze(0) * vq
            v[:, q_idx] = s.unsqueeze(0) * vp + c.unsqueeze(0) * vq
            ring = [ring[-1]] + ring[:-1]
    w = work.diagonal().clone()
    order = torch.argsort(w)
    return w[order].to(a.dtype), v[:, order].to(a.dtype)


def eig_selective(a: torch.Tensor, n_eig: int, largest: bool = True,
                  method: str = "auto", tol: float = 1e-7):
    """Selective eigendecomposition (detail/eig.cuh:174-217 syevdx analog):
    top/bottom n_eig pairs WITHOUT the full decomposition when profitable.

    method="lobpcg" (auto for n_eig << n): blocked LOBPCG iteration — cost
    O(n^2 * n_eig) per step vs syevd's O(n^3); method="full" falls back to
    eigh + slice (the reference's syevdx also degenerates to full cost for
    wide ranges).
    """
    n = a.shape[0]
    use_lobpcg = method == "lobpcg" or (method == "auto"
                                        and 0 < n_eig <= max(1, n // 8)
                                        and n >= 64)
    if use_lobpcg:
        try:
            af = a.double()
            w, v = torch.lobpcg(af if largest else -af, k=n_eig,
                                largest=True, tol=tol)
            if not largest:
                w = -w
            order = torch.argsort(w)
            return w[order].to(a.dtype), v[:, order].to(a.dtype)
        except Exception:
            pass  # LOBPCG can fail on clustered spectra: full fallback
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
