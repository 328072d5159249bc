"""Batched top-k / bottom-k selection (the k-selection engine).

Reference parity: raft/matrix/select_k.cuh:75, detail/select_k-inl.cuh
(learned dispatch tree :38-65), detail/select_radix.cuh (multi-pass MSB radix,
AIR top-k) and detail/select_warpsort.cuh (warp bitonic priority queues).

MI355X design (csrc/select_k.hip): the warpsort family is re-derived for
64-wide wavefronts — per-lane sorted register queues of Capacity/64 elements
merged with wave-wide bitonic exchanges (vs the reference's 32-lane queues);
the radix path uses 8-bit digits with LDS histograms + device-scope atomic
merge and the candidate-compaction buffer trick. The dispatch heuristic
(k<=wave-queue capacity -> warpsort; else radix) is re-measured on gfx950
rather than copying the reference's learned tree (select_k-inl.cuh:38-65 was
trained on NVIDIA parts).
"""
from __future__ import annotations

from enum import Enum

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu


class SelectAlgo(Enum):
    AUTO = "auto"
    RADIX = "radix"
    WARPSORT = "warpsort"
    TORCH = "torch"   # vendor topk (used for cross-checking, like the cub-sort check)


def select_k(x: torch.Tensor, k: int, select_min: bool = True,
             algo: SelectAlgo = SelectAlgo.AUTO, sorted: bool = True):
    """Per-row k smallest (or largest) values of a [batch, len] matrix.

    Returns (values [batch,k], indices [batch,k] int64).

    NaN semantics: both native engines canonicalize NaN (any sign/payload) to
    the maximum ordinal, so NaN orders after every finite value and +/-inf in
    BOTH selection directions; NaNs are only selected when a row has fewer
    than k non-NaN values, and then they carry their real in-range indices.
    """
    assert x.dim() == 2
    batch, n = x.shape
    k = int(k)
    assert 0 < k <= n, f"k={k} out of range for row length {n}"

    if (on_gpu(x) and algo != SelectAlgo.TORCH
            and x.dtype in (torch.bfloat16, torch.float16) and k <= 2048
            and n * batch < (1 << 31)):
        # half dtypes at small k: widen once to fp32 and ride the fp32
        # radix/warpsort engines — one extra 2x write, vs the generic
        # engine's histogram + 2 filter re-reads (measured [8192 x 100k]
        # k=64 bf16: generic 5.5 ms / torch.topk 3.7 -> this route 3.35).
        # bf16->fp32 is exact, so selection and values are unchanged.
        vals, idx = select_k(x.float(), k, select_min=select_min, algo=algo,
                             sorted=sorted)
        return vals.to(x.dtype), idx

    if (on_gpu(x) and algo != SelectAlgo.TORCH
            and (x.dtype in (torch.float64, torch.bfloat16, torch.float16)
                 or (x.dtype == torch.float32 and k > 2048))):
        # generic native engine: any dtype (64-bit ordinals for fp64),
        # unbounded k, int64 indices (rows may exceed 2^31 elements).
        # Output is unsorted; sort the [batch, k] slab here when asked.
        ext = require_ext()
        vals, idx = ext.select_k_generic(x.contiguous(), None, k=k,
                                         select_min=bool(select_min))
        if sorted:
            order = torch.argsort(vals, dim=1, descending=not select_min)
            vals = torch.gather(vals, 1, order)
            idx = torch.gather(idx, 1, order)
        return vals, idx

    if (on_gpu(x) and x.dtype == torch.float32 and algo != SelectAlgo.TORCH
            and k <= 64 and n <= 4096
            and algo in (SelectAlgo.AUTO, SelectAlgo.WARPSORT)):
        # short rows, small k: the wave-register warpsort queue beats both
        # rocPRIM topk and the radix path (measured [20000 x 500] k=32:
        # warpsort 0.108 ms vs torch.topk 0.27 ms vs radix 0.43 ms)
        ext = require_ext()
        vals, idx = ext.select_k(x.contiguous(), k, bool(select_min), 2,
                                 bool(sorted))
        return vals, idx.to(torch.int64)

    if (on_gpu(x) and x.dtype == torch.float32 and algo != SelectAlgo.TORCH
            and k <= 2048 and n > 4096):
        # remaining n <= 4096 shapes (k > 64) are cheapest through the vendor
        # segmented sort (rocPRIM topk) — the same shape-dispatch idea as the
        # reference's learned tree, re-measured on gfx950 (see benchmarks)
        ext = require_ext()
        # two-level split: a small batch over a huge row leaves the chip idle
        # (one workgroup per row); split rows into S segments, select per
        # segment, then select over the S*k candidates (the same multi-block
        # merge idea as the reference's two-pass warpsort).
        if batch < 256 and n >= 262144 and n >= 8 * k:
            s = 1
            while batch * s * 2 <= 2048 and (n // (s * 2)) >= max(4 * k, 4096):
                s *= 2
            if s > 1:
                seg = -(-n // s)          # ceil
                pad = seg * s - n
                if pad:
                    fill = float("inf") if select_min else float("-inf")
                    xp = torch.cat([x, torch.full((batch, pad), fill, dtype=x.dtype,
                                                  device=x.device)], dim=1)
                else:
                    xp = x
                segs = xp.reshape(batch * s, seg).contiguous()
                sv, si = ext.select_k(segs, min(k, seg), bool(select_min), 0, False)
                kk = sv.shape[1]
                cand_v = sv.reshape(batch, s * kk)
                base = (torch.arange(s, device=x.device, dtype=torch.int64) * seg)
                cand_i = (si.to(torch.int64).reshape(batch, s, kk)
                          + base.view(1, s, 1)).reshape(batch, s * kk)
                fv, fpos = ext.select_k(cand_v.contiguous(), k, bool(select_min),
                                        0, bool(sorted))
                gi = torch.gather(cand_i, 1, fpos.to(torch.int64))
                # padded +/-inf slots carry fabricated indices base+pos >= n;
                # they are only selected when a row has < k real candidates —
                # clamp so callers never see an out-of-range index
                gi.clamp_(max=n - 1)
                return fv, gi
        algo_code = {SelectAlgo.AUTO: 0, SelectAlgo.RADIX: 1, SelectAlgo.WARPSORT: 2}[algo]
        if algo == SelectAlgo.AUTO and k <= 64 and batch >= 2048 and n >= 100000:
            # measured on gfx950 (benchmarks 2026-09): the wave-register
            # warpsort queue is ~6x the radix path on long rows once the
            # batch fills the chip (it degenerates to a streaming read after
            # the ballot filter warms up); radix wins on shorter rows / small
            # batch. This is the re-measured analog of the reference's
            # learned dispatch tree.
            algo_code = 2
        vals, idx = ext.select_k(x.contiguous(), k, bool(select_min), algo_code, bool(sorted))
        return vals, idx.to(torch.int64)

    vals, idx = torch.topk(x, k, dim=1, largest=not select_min, sorted=sorted)
    return vals, idx
