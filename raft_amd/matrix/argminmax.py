"""Row-wise argmin/argmax (reference: raft/matrix/argmin.cuh, argmax.cuh —
coalesced_reduction with a key-value-pair argmin op).

On GPU the fused L2-NN kernel covers the hot case; standalone argmin over an
arbitrary matrix uses the native rowwise argmin kernel (csrc/reductions.hip)
for fp32, else the vendor reduction.
"""
from __future__ import annotations

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu


def argmin(x: torch.Tensor) -> torch.Tensor:
    """Per-row argmin (reference matrix::argmin)."""
    assert x.dim() == 2
    if on_gpu(x) and x.dtype == torch.float32:
        ext = require_ext()
        return ext.row_argmin(x.contiguous()).to(torch.int64)
    return x.argmin(dim=1)


def argmax(x: torch.Tensor) -> torch.Tensor:
    """Per-row argmax (reference matrix::argmax)."""
    assert x.dim() == 2
    if on_gpu(x) and x.dtype == torch.float32:
        ext = require_ext()
        return ext.row_argmin(x.neg().contiguous()).to(torch.int64)
    return x.argmax(dim=1)
