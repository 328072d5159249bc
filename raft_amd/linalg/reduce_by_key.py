"""Keyed reductions — the k-means centroid-update primitive.

Reference parity: raft/linalg/reduce_rows_by_key (detail, 4 kernels incl.
smem-binned) and reduce_cols_by_key (atomics).

MI355X design: on GPU, reduce_rows_by_key runs the native kernel in
csrc/kmeans.hip — per-workgroup LDS accumulation when n_keys*D fits in the
160 KiB LDS, else direct fp32 device-scope atomics into the output (Guideline 12:
per-block pre-aggregation first). CPU path uses torch.index_add_ (the oracle).
"""
from __future__ import annotations

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu


def reduce_rows_by_key(x: torch.Tensor, keys: torch.Tensor, n_keys: int | None = None,
                       weights: torch.Tensor | None = None,
                       out_dtype: torch.dtype | None = None) -> torch.Tensor:
    """sums[k, :] = sum over rows i with keys[i]==k of (w_i *) x[i, :].

    out_dtype: accumulation/output dtype. Defaults to fp32 for half-precision
    inputs (bf16 cannot represent sums of >256 unit-scale terms exactly) and
    x.dtype otherwise.
    """
    assert x.dim() == 2 and keys.dim() == 1 and keys.shape[0] == x.shape[0]
    if n_keys is None:
        n_keys = int(keys.max().item()) + 1 if keys.numel() else 0
    if out_dtype is None:
        out_dtype = torch.float32 if x.dtype in (torch.bfloat16, torch.float16) \
            else x.dtype
    if on_gpu(x, keys) and x.dtype == torch.float32 and weights is None \
            and out_dtype == torch.float32:
        ext = require_ext()
        if x.shape[0] >= 65536:
            # sort-based: atomics only at run boundaries (the naive atomic
            # kernel is issue-rate bound — measured 33 ms @ 10M x 256)
            k32 = keys.to(torch.int32)
            keys_sorted, perm = torch.sort(k32)
            return ext.reduce_rows_by_key_sorted(x.contiguous(),
                                                 perm.to(torch.int32).contiguous(),
                                                 keys_sorted.contiguous(), int(n_keys))
        return ext.reduce_rows_by_key(x.contiguous(), keys.to(torch.int32).contiguous(), int(n_keys))
    # weighted variant (reference has one): torch index_add over w*x
    out = torch.zeros((n_keys, x.shape[1]), dtype=out_dtype, device=x.device)
    src = x if weights is None else x * weights.unsqueeze(1)
    out.index_add_(0, keys.to(torch.int64), src.to(out_dtype))
    return out


def reduce_cols_by_key(x: torch.Tensor, keys: torch.Tensor, n_keys: int | None = None) -> torch.Tensor:
    """out[:, k] = sum over cols j with keys[j]==k of x[:, j]."""
    assert x.dim() == 2 and keys.dim() == 1 and keys.shape[0] == x.shape[1]
    if n_keys is None:
        n_keys = int(keys.max().item()) + 1 if keys.numel() else 0
    out = torch.zeros((x.shape[0], n_keys), dtype=x.dtype, device=x.device)
    out.index_add_(1, keys.to(torch.int64), x)
    return out
