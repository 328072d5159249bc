"""BLAS-level ops: gemm/gemv/dot/axpy + fp32-emulated GEMM for MI355X.

Reference parity: raft/linalg/gemm.cuh:51-169, gemv.hpp, dot, axpy (cuBLAS
wrappers; cublas_wrappers.hpp).

MI355X design: plain GEMMs go to rocBLAS/hipBLASLt via torch.matmul (that IS
the vendor path on ROCm). CDNA4 has **no fp32-input MFMA** — native SGEMM runs
on the 157 TF vector ALU. For fp32 workloads that are GEMM-shaped (pairwise
distance, k-means, PCA covariance) we therefore provide *split-bf16 fp32
emulation* on the 2.5 PF bf16 matrix cores:

    a = a_hi + a_mid + a_lo   (three bf16 slices capture fp32's 24-bit mantissa)
    A@B ≈ Σ products of slices, accumulated in fp32 (rocblas_gemm_ex bf16-in/f32-out)

``bf16x3`` (6 slice-products) reproduces fp32-level accuracy (validated against
fp64 in tests/test_gemm.py); ``bf16x2`` (3 products) is TF32-class. This is the
Ootomo-Yokota-style emulation scheme, re-derived for CDNA4's bf16 MFMA rates.
"""
from __future__ import annotations

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu


def gemm(a: torch.Tensor, b: torch.Tensor, alpha: float = 1.0, beta: float = 0.0,
         c: torch.Tensor | None = None, trans_a: bool = False, trans_b: bool = False) -> torch.Tensor:
    """C = alpha * op(A) @ op(B) + beta * C (rocBLAS via torch)."""
    if trans_a:
        a = a.t()
    if trans_b:
        b = b.t()
    if (a.is_cuda and a.dim() == 2 and b.dim() == 2
            and a.shape[0] * b.shape[1] > (1 << 30) and a.shape[0] > 1):
        # vendor GEMM corrupts outputs >= 2^31 elements on this stack
        # (32-bit C-element indexing, measured — see BASELINE.md): row-chunk
        m, n = a.shape[0], b.shape[1]
        rows = max(1, (1 << 30) // n)
        out = torch.empty((m, n), dtype=a.dtype, device=a.device)
        for r0 in range(0, m, rows):
            torch.matmul(a[r0:r0 + rows], b, out=out[r0:r0 + rows])
    else:
        out = torch.matmul(a, b)
    if alpha != 1.0:
        out = out * alpha
    if c is not None and beta != 0.0:
        out = out + beta * c
    return out


def gemv(a: torch.Tensor, x: torch.Tensor, alpha: float = 1.0, beta: float = 0.0,
         y: torch.Tensor | None = None, trans: bool = False) -> torch.Tensor:
    """y = alpha*A@x + beta*y (rocBLAS gemv analog)."""
    m = a.t() if trans else a
    out = torch.mv(m, x) * alpha
    if y is not None and beta != 0.0:
        out = out + beta * y
    return out


def dot(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """BLAS-1 dot product (rocBLAS via torch)."""
    return torch.dot(x.reshape(-1), y.reshape(-1))


def axpy(alpha: float, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """y += alpha * x (in-place on y)."""
    y.add_(x, alpha=alpha)
    return y


# ---------------------------------------------------------------------------
# fp32 emulation on bf16 MFMA
# ---------------------------------------------------------------------------

def _split_bf16(a: torch.Tensor, n: int):
    """Split fp32 tensor into n bf16 slices with a ≈ sum(slices)."""
    slices = []
    resid = a
    for _ in range(n):
        s = resid.to(torch.bfloat16)
        slices.append(s)
        resid = resid - s.to(torch.float32)
    return slices


def gemm_bf16_f32(a_bf16: torch.Tensor, b_bf16: torch.Tensor,
                  out: torch.Tensor | None = None, beta: float = 0.0) -> torch.Tensor:
    """bf16 x bf16 -> fp32 GEMM with fp32 accumulate (rocblas_gemm_ex).

    On CPU this is emulated with fp32 math (the oracle); on GPU it calls the
    extension's rocBLAS wrapper — torch.matmul would round the output to bf16,
    which destroys the split-emulation scheme.
    """
    if on_gpu(a_bf16, b_bf16):
        ext = require_ext()
        return ext.gemm_bf16_f32(a_bf16.contiguous(), b_bf16.contiguous(), out, float(beta))
    res = torch.matmul(a_bf16.to(torch.float32), b_bf16.to(torch.float32))
    if out is not None:
        if beta != 0.0:
            out.mul_(beta).add_(res)
        else:
            out.copy_(res)
        return out
    return res


def gemm_fp32_emulated(a: torch.Tensor, b: torch.Tensor, mode: str = "bf16x3") -> torch.Tensor:
    """fp32 GEMM on bf16 matrix cores via mantissa splitting.

    mode "bf16x3": 6 slice-products -> fp32-class accuracy (~2^-24 rel).
    mode "bf16x2": 3 slice-products -> TF32-class accuracy (~2^-16 rel).
    """
    assert a.dtype == torch.float32 and b.dtype == torch.float32
    fused = on_gpu(a, b)  # rocBLAS beta=1 accumulates in-GEMM (no extra passes)
    if mode == "bf16x2":
        ah, al = _split_bf16(a, 2)
        bh, bl = _split_bf16(b, 2)
        c = gemm_bf16_f32(ah, bh)
        terms = [(ah, bl), (al, bh)]
    elif mode == "bf16x3":
        ah, am, al = _split_bf16(a, 3)
        bh, bm, bl = _split_bf16(b, 3)
        c = gemm_bf16_f32(ah, bh)
        terms = [(ah, bm), (am, bh), (am, bm), (ah, bl), (al, bh)]
    elif mode == "native":
        return torch.matmul(a, b)
    else:
        raise ValueError(f"unknown fp32 emulation mode {mode}")
    for ta, tb in terms:
        if fused:
            gemm_bf16_f32(ta, tb, out=c, beta=1.0)
        else:
            c += gemm_bf16_f32(ta, tb)
    return c
