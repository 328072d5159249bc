"""Pairwise distances — the contraction-engine workload.

Reference parity: RAFT's historical raft/distance tree (pairwise_distance with
L2 expanded/unexpanded, cosine, L1, Linf, Lp, Hamming, ...) built on the tiled
contraction engine (linalg/contractions.cuh) — required by BASELINE.json
although the post-cuVS-split snapshot dropped the tree.

MI355X design: CDNA4 has no fp32 MFMA, so the expanded-form distances
(L2-expanded, cosine, inner product) are GEMM-dominated and run through either
  * rocBLAS SGEMM (fp32 vector ALU, ~157 TF ceiling), or
  * split-bf16 emulation on the 2.5 PF bf16 matrix cores
    (linalg.gemm_fp32_emulated; fp32-class accuracy with mode="bf16x3"),
with the norm/epilogue (||x||^2 + ||y||^2 - 2xy, clamp, sqrt) fused into ONE
HIP kernel (csrc/pairwise.hip) so the M x N distance tile is touched once.
Unexpanded forms (L1/Linf/Lp/Hamming/Canberra...) are diff-accumulation
contractions; on GPU they run a tiled LDS-staged kernel for fp32.
"""
from __future__ import annotations

from enum import Enum

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu
from raft_amd.linalg.gemm import gemm_fp32_emulated


class DistanceType(Enum):
    L2Expanded = "sqeuclidean"          # squared L2 via expansion
    L2SqrtExpanded = "euclidean"        # sqrt of the above
    L2Unexpanded = "sqeuclidean_unexp"
    L2SqrtUnexpanded = "euclidean_unexp"
    InnerProduct = "inner_product"
    CosineExpanded = "cosine"
    L1 = "l1"
    Linf = "linf"
    LpUnexpanded = "lp"
    Canberra = "canberra"
    HammingUnexpanded = "hamming"
    JensenShannon = "jensenshannon"
    KLDivergence = "kl_divergence"
    CorrelationExpanded = "correlation"
    RusselRaoExpanded = "russelrao"


_SQRT_TYPES = {DistanceType.L2SqrtExpanded, DistanceType.L2SqrtUnexpanded}
_EXpanded_L2 = {DistanceType.L2Expanded, DistanceType.L2SqrtExpanded,
                DistanceType.L2Unexpanded, DistanceType.L2SqrtUnexpanded}


def pairwise_distance(x: torch.Tensor, y: torch.Tensor | None = None,
                      metric: DistanceType | str = DistanceType.L2Expanded,
                      p: float = 2.0, fp32_mode: str = "auto") -> torch.Tensor:
    """Dense pairwise distance matrix [m, n] between rows of x [m,d] and y [n,d].

    fp32_mode: "auto" | "native" | "bf16x3" | "bf16x2" — GEMM engine for the
    expanded forms on fp32 inputs (see linalg.gemm_fp32_emulated).
    """
    if isinstance(metric, str):
        metric = _metric_from_str(metric)
    if y is None:
        y = x
    assert x.dim() == 2 and y.dim() == 2 and x.shape[1] == y.shape[1]

    if metric in _EXpanded_L2:
        d2 = _l2_squared(x, y, fp32_mode)
        return d2.sqrt() if metric in _SQRT_TYPES else d2
    if metric == DistanceType.InnerProduct:
        return _gemm_xyt(x, y, fp32_mode)
    if metric == DistanceType.CosineExpanded:
        xn = torch.nn.functional.normalize(x, dim=1, eps=1e-12)
        yn = torch.nn.functional.normalize(y, dim=1, eps=1e-12)
        return (1.0 - _gemm_xyt(xn, yn, fp32_mode)).clamp_min(0.0)
    if metric == DistanceType.CorrelationExpanded:
        xc = x - x.mean(dim=1, keepdim=True)
        yc = y - y.mean(dim=1, keepdim=True)
        xn = torch.nn.functional.normalize(xc, dim=1, eps=1e-12)
        yn = torch.nn.functional.normalize(yc, dim=1, eps=1e-12)
        return (1.0 - _gemm_xyt(xn, yn, fp32_mode)).clamp_min(0.0)

    # unexpanded (diff-accumulation) forms
    if on_gpu(x, y) and x.dtype == torch.float32 and metric in (
            DistanceType.L1, DistanceType.Linf, DistanceType.LpUnexpanded,
            DistanceType.Canberra, DistanceType.HammingUnexpanded):
        ext = require_ext()
        code = {DistanceType.L1: 0, DistanceType.Linf: 1, DistanceType.LpUnexpanded: 2,
                DistanceType.Canberra: 3, DistanceType.HammingUnexpanded: 4}[metric]
        return ext.pairwise_unexpanded(x.contiguous(), y.contiguous(), code, float(p))
    return _unexpanded_ref(x, y, metric, p)


def _metric_from_str(s: str) -> DistanceType:
    for m in DistanceType:
        if m.value == s or m.name.lower() == s.lower():
            return m
    raise ValueError(f"unknown metric {s!r}")


def _gemm_xyt(x, y, fp32_mode):
    if x.dtype == torch.float32 and fp32_mode not in ("native", "auto"):
        return gemm_fp32_emulated(x, y.t(), mode=fp32_mode)
    m, n = x.shape[0], y.shape[0]
    if x.is_cuda and m * n > (1 << 30) and m > 1:
        # vendor GEMM (hipBLASLt via torch.matmul, rocblas_gemm_ex) corrupts
        # outputs >= 2^31 elements on this stack (measured: [256 x 30M] fp32
        # C exact for rows 0..~200, garbage near the tail — 32-bit C-element
        # indexing). Row-chunk below the boundary; the
        # raft_amd HIP kernels index 64-bit and are unaffected.
        rows = max(1, (1 << 30) // n)
        out = torch.empty((m, n), dtype=x.dtype, device=x.device)
        for r0 in range(0, m, rows):
            torch.matmul(x[r0:r0 + rows], y.t(), out=out[r0:r0 + rows])
        return out
    return x @ y.t()


def _l2_squared(x, y, fp32_mode):
    """||x||^2 + ||y||^2 - 2 x.y with fused epilogue on GPU."""
    if (on_gpu(x, y) and x.shape[1] % 64 != 0
            and (x.dtype == torch.bfloat16
                 or (x.dtype == torch.float32
                     and fp32_mode in ("bf16x2", "bf16x3", "mfma")))):
        # zero feature columns change no L2 distance: pad to the MFMA K
        # granularity so any d rides the single-write tile kernel
        dp = (-x.shape[1]) % 64
        return _l2_squared(torch.nn.functional.pad(x, (0, dp)),
                           torch.nn.functional.pad(y, (0, dp)), fp32_mode)
    if on_gpu(x, y) and x.dtype == torch.bfloat16:
        ext = require_ext()
        xn = x.float().pow(2).sum(dim=1)
        yn = y.float().pow(2).sum(dim=1)
        if x.shape[1] % 64 == 0:
            # single-write MFMA tile kernel (epilogue fused into the C-write)
            return ext.pairwise_l2_mfma([x.contiguous()], [y.contiguous()],
                                        xn.contiguous(), yn.contiguous())
        g = ext.gemm_bf16_f32_nt(x.contiguous(), y.contiguous())
        return ext.l2_epilogue_(g, xn.contiguous(), yn.contiguous())
    if (on_gpu(x, y) and x.dtype == torch.float32 and x.shape[1] % 64 == 0
            and fp32_mode in ("bf16x2", "bf16x3", "mfma")):
        # split-bf16 MFMA tile kernel: distance tile written exactly once
        from raft_amd.neighbors.fused_l2nn import split_bf16_slices
        ext = require_ext()
        nsl = 2 if fp32_mode == "bf16x2" else 3
        xs = split_bf16_slices(x, nsl)
        ys = split_bf16_slices(y, nsl)
        xn = (x * x).sum(dim=1)
        yn = (y * y).sum(dim=1)
        return ext.pairwise_l2_mfma(xs, ys, xn.contiguous(), yn.contiguous())
    xn = (x.double() * x.double()).sum(dim=1) if x.device.type == "cpu" else (x * x).sum(dim=1)
    yn = (y.double() * y.double()).sum(dim=1) if y.device.type == "cpu" else (y * y).sum(dim=1)
    g = _gemm_xyt(x, y, fp32_mode)
    if on_gpu(x, y) and x.dtype == torch.float32:
        ext = require_ext()
        # fused: d2 = xn[:,None] + yn[None,:] - 2 g, clamped at 0, in one pass
        return ext.l2_epilogue_(g, xn.to(torch.float32).contiguous(),
                                yn.to(torch.float32).contiguous())
    d2 = xn.unsqueeze(1) + yn.unsqueeze(0) - 2.0 * g.double()
    return d2.clamp_min(0).to(x.dtype)


def _unexpanded_ref(x, y, metric, p):
    xd, yd = x.double(), y.double()
    diff = xd.unsqueeze(1) - yd.unsqueeze(0)  # [m, n, d] — reference path only
    if metric == DistanceType.L1:
        out = diff.abs().sum(dim=2)
    elif metric == DistanceType.Linf:
        out = diff.abs().max(dim=2).values
    elif metric == DistanceType.LpUnexpanded:
        out = diff.abs().pow(p).sum(dim=2).pow(1.0 / p)
    elif metric == DistanceType.Canberra:
        denom = xd.abs().unsqueeze(1) + yd.abs().unsqueeze(0)
        out = torch.where(denom > 0, diff.abs() / denom, torch.zeros_like(denom)).sum(dim=2)
    elif metric == DistanceType.HammingUnexpanded:
        out = (diff != 0).double().mean(dim=2)
    elif metric == DistanceType.KLDivergence:
        xe = xd.unsqueeze(1).clamp_min(1e-300)
        ye = yd.unsqueeze(0).clamp_min(1e-300)
        out = (xe * (xe / ye).log()).sum(dim=2)
    elif metric == DistanceType.JensenShannon:
        xe = xd.unsqueeze(1).clamp_min(0)
        ye = yd.unsqueeze(0).clamp_min(0)
        m = 0.5 * (xe + ye)
        def _kl(a, b):
            r = torch.where(a > 0, a * (a.clamp_min(1e-300) / b.clamp_min(1e-300)).log(),
                            torch.zeros_like(a))
            return r.sum(dim=2)
        out = torch.sqrt(0.5 * _kl(xe, m) + 0.5 * _kl(ye, m))
    elif metric == DistanceType.RusselRaoExpanded:
        d = x.shape[1]
        out = (d - (xd.unsqueeze(1) * yd.unsqueeze(0)).sum(dim=2)) / d
    else:
        raise ValueError(metric)
    return out.to(x.dtype)
