"""Shared helpers: dispatch, dtype utilities, chunking for 288 GB HBM sizing."""
from __future__ import annotations

import math

import torch

from raft_amd._ext import require_ext, ext_or_none, has_ext  # re-export

__all__ = [
    "require_ext", "ext_or_none", "has_ext", "on_gpu", "check_same_device",
    "row_chunks", "as_2d", "torch_dtype_name",
]


def on_gpu(*tensors: torch.Tensor) -> bool:
    """True when all tensors live on a HIP device (dispatch to native kernels)."""
    return all(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


def check_same_device(*tensors: torch.Tensor) -> torch.device:
    devs = {t.device for t in tensors if isinstance(t, torch.Tensor)}
    if len(devs) != 1:
        raise ValueError(f"tensors must be co-located; got devices {devs}")
    return next(iter(devs))


def row_chunks(n_rows: int, max_rows: int):
    """Yield (start, end) row ranges of at most max_rows."""
    for s in range(0, n_rows, max_rows):
        yield s, min(s + max_rows, n_rows)


def as_2d(t: torch.Tensor) -> torch.Tensor:
    return t.unsqueeze(0) if t.dim() == 1 else t


def torch_dtype_name(dtype: torch.dtype) -> str:
    return str(dtype).replace("torch.", "")


def ceil_div(a: int, b: int) -> int:
    return -(-a // b)


def next_pow2(x: int) -> int:
    return 1 if x <= 1 else 2 ** math.ceil(math.log2(x))
