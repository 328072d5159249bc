"""Fused L2 nearest-neighbor (distance + argmin without materialization).

Reference parity: RAFT's historical fused-L2-NN (the contraction engine with a
key-value argmin epilogue) — required by BASELINE.json config 5 and the
k-means EM loop.

MI355X design (csrc/fused_l2nn.hip): each workgroup owns a 128-row tile of X
and loops over Y in 128-column tiles; the running (min, argmin) pair per row
lives in registers, so the m x n distance matrix never touches HBM. fp32
inputs are pre-split into bf16 slices and the dot term runs on the 2.5 PF
bf16 matrix cores:
    nslice=2 (fp32 mode "bf16x2"): 3 MFMA slice-products, ~2^-16 accuracy
    nslice=3 (fp32 mode "bf16x3"): 6 products, fp32-class accuracy
bf16 inputs run nslice=1 directly. A rocBLAS-GEMM + epilogue chunked path
remains as the fallback for shapes the fused kernel does not cover
(d % 64 != 0) and for the explicit "native" fp32 engine.
"""
from __future__ import annotations

import torch

from raft_amd._ext import ext_or_none, require_ext
from raft_amd.utils import on_gpu, row_chunks
from raft_amd.linalg.gemm import gemm_fp32_emulated

_MODE_NSLICE = {"bf16x2": 2, "bf16x3": 3, "bf16x2v": 2, "bf16x1v": 1,
                "auto": 2, "fused": 3}
#: modes that run the exact-fp32 verification/repair pass (provably exact
#: argmin: rows inside the split-error margin are rescanned in fp32, and the
#: chosen distance is recomputed exactly for every row)
_VERIFY_MODES = {"bf16x2v", "bf16x1v", "auto"}
#: (lead, tail) margin-bound constants per verified mode — the provable
#: |Δdot| <= lead*sqrt(xn*cn) + tail*(xn+cn) envelope of the split emulation
#: (derivation: csrc/kmeans.hip l2nn_verify_repair_kernel). bf16x1v does a
#: single MFMA product (1/3 the matrix work of bf16x2v) against a 2^6-wider
#: bound; correctness is identical — only the rescan fraction grows on data
#: with near-tied neighbors.
_MODE_BOUND = {"bf16x1v": (2.0 ** -7, 2.0 ** -12)}
_DEFAULT_BOUND = (2.0 ** -13, 2.0 ** -18)


def split_bf16_slices(t: torch.Tensor, nslice: int):
    """fp32 -> bf16 slices with t ≈ sum(slices). Iteration-invariant for the
    k-means X matrix, so callers may precompute (see kmeans_iterate)."""
    slices = []
    resid = t
    for i in range(nslice):
        s = resid.to(torch.bfloat16)
        slices.append(s.contiguous())
        if i + 1 < nslice:
            resid = resid - s.to(torch.float32)
    return slices


def _pad_cols(y: torch.Tensor, yn: torch.Tensor, mult: int = 128):
    n = y.shape[0]
    pad = (-n) % mult
    if pad == 0:
        return y, yn, n
    yp = torch.cat([y, torch.zeros(pad, y.shape[1], dtype=y.dtype, device=y.device)])
    ynp = torch.cat([yn, torch.full((pad,), float("inf"), dtype=yn.dtype, device=yn.device)])
    return yp, ynp, n


def fused_l2nn_presplit(x_slices, xn: torch.Tensor, y: torch.Tensor,
                        sqrt: bool = False, int32_labels: bool = False,
                        verify_x: torch.Tensor | None = None,
                        bound: tuple[float, float] = _DEFAULT_BOUND):
    """Fused kernel entry with precomputed X slices (k-means hot loop).

    verify_x: the original fp32 matrix — enables the exact-fp32
    verification/repair pass (csrc/kmeans.hip l2nn_verify_repair): every row's
    chosen distance is recomputed in exact fp32, and rows whose
    (best, second-best) margin falls inside the provable split-emulation error
    bound rescan all centroids exactly. Result: fp32-exact argmin + fp32
    distances at split-bf16 MFMA speed.
    """
    ext = require_ext()
    nslice = len(x_slices)
    yn = (y * y).sum(dim=1)
    yp, ynp, n_true = _pad_cols(y, yn)
    y_slices = split_bf16_slices(yp, nslice)
    dmin, amin, dmin2 = ext.fused_l2nn_split(list(x_slices), list(y_slices),
                                             xn.contiguous(), ynp.contiguous())
    if verify_x is not None:
        cn_max = yn.max().reshape(1)   # device scalar: no host sync
        ext.l2nn_verify_repair(verify_x.contiguous(), y.contiguous(),
                               xn.contiguous(), dmin, amin, dmin2, cn_max,
                               lead=bound[0], tail=bound[1])
    if sqrt:
        dmin = dmin.clamp_min(0).sqrt()
    return dmin, (amin if int32_labels else amin.to(torch.int64))


def fused_l2nn(x: torch.Tensor, y: torch.Tensor, sqrt: bool = False,
               fp32_mode: str = "auto", chunk_rows: int = 262144):
    """For each row of x [m,d]: (min L2 distance to rows of y [n,d], argmin).

    Returns (min_dists [m], argmins [m] int64). Distances are squared L2
    unless sqrt=True.
    """
    assert x.dim() == 2 and y.dim() == 2 and x.shape[1] == y.shape[1]
    d = x.shape[1]

    if on_gpu(x, y):
        if x.dtype == torch.float16:
            # exact widening -> the fp32 engines (incl. verified modes)
            dmin, amin = fused_l2nn(x.float(), y.float(), sqrt=sqrt,
                                    fp32_mode=fp32_mode,
                                    chunk_rows=chunk_rows)
            return dmin, amin
        if (d % 64 != 0 and x.dtype in (torch.float32, torch.bfloat16)
                and (x.dtype == torch.bfloat16 or fp32_mode in _MODE_NSLICE)):
            # zero columns leave every pairwise distance unchanged: pad to
            # the MFMA K granularity and take the fused path (closes the
            # round-1 "d % 64 falls back to chunked GEMM" perf cliff)
            dp = (-d) % 64
            xp = torch.nn.functional.pad(x, (0, dp))
            yp = torch.nn.functional.pad(y, (0, dp))
            return fused_l2nn(xp, yp, sqrt=sqrt, fp32_mode=fp32_mode,
                              chunk_rows=chunk_rows)
        if x.dtype == torch.bfloat16 and d % 64 == 0:
            ext = require_ext()
            xn = x.to(torch.float32).pow(2).sum(dim=1)
            yf = y.to(torch.float32)
            yn = (yf * yf).sum(dim=1)
            yp, ynp, _ = _pad_cols(y.contiguous(), yn)
            dmin, amin, _ = ext.fused_l2nn_split([x.contiguous()], [yp.contiguous()],
                                                 xn.contiguous(), ynp.contiguous())
            if sqrt:
                dmin = dmin.clamp_min(0).sqrt()
            return dmin, amin.to(torch.int64)
        if x.dtype == torch.float32:
            if fp32_mode in _MODE_NSLICE and d % 64 == 0:
                nslice = _MODE_NSLICE[fp32_mode]
                xs = split_bf16_slices(x, nslice)
                xn = (x * x).sum(dim=1)
                vx = x if fp32_mode in _VERIFY_MODES else None
                return fused_l2nn_presplit(
                    xs, xn, y, sqrt=sqrt, verify_x=vx,
                    bound=_MODE_BOUND.get(fp32_mode, _DEFAULT_BOUND))
            return _chunked_gpu(x, y, sqrt, fp32_mode, chunk_rows)

    # CPU oracle
    d2 = torch.cdist(x.double(), y.double(), p=2) ** 2
    dmin, amin = d2.min(dim=1)
    if sqrt:
        dmin = dmin.clamp_min(0).sqrt()
    return dmin.to(x.dtype), amin


def _chunked_gpu(x, y, sqrt, fp32_mode, chunk_rows):
    """GEMM + fused argmin-epilogue, chunked so the distance tile stays small."""
    ext = require_ext()
    m = x.shape[0]
    # keep each GEMM output under 2^30 elements: vendor GEMM corrupts
    # >= 2^31-element outputs on this stack (32-bit C-element indexing,
    # measured — BASELINE.md)
    chunk_rows = max(1, min(chunk_rows, (1 << 30) // max(y.shape[0], 1)))
    yn = (y * y).sum(dim=1)
    xn = (x * x).sum(dim=1)
    dmin = torch.empty(m, dtype=x.dtype, device=x.device)
    amin = torch.empty(m, dtype=torch.int32, device=x.device)
    for s, e in row_chunks(m, chunk_rows):
        g = _gemm_xyt(x[s:e], y, fp32_mode)
        # fused: per row argmin of (xn + yn - 2g) — never materializes d2
        ext.l2nn_epilogue(g, xn[s:e].contiguous(), yn.contiguous(),
                          dmin[s:e], amin[s:e])
    if sqrt:
        dmin = dmin.clamp_min(0).sqrt()
    return dmin, amin.to(torch.int64)


def _gemm_xyt(x, y, fp32_mode):
    # _chunked_gpu clamps chunk_rows so the output stays < 2^30 elements
    # (vendor GEMM 32-bit C-index overflow); the emulated
    # path's rocblas wrappers additionally self-chunk.
    if x.dtype == torch.float32 and fp32_mode in ("bf16x3", "bf16x2"):
        return gemm_fp32_emulated(x, y.t(), mode=fp32_mode)
    return x @ y.t()


def fused_l2nn_argmin(x: torch.Tensor, y: torch.Tensor, **kw) -> torch.Tensor:
    """Argmin-only convenience over fused_l2nn (reference fusedL2NNArgmin)."""
    return fused_l2nn(x, y, **kw)[1]
