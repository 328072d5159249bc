"""Row/column reductions.

Reference parity: raft/linalg/reduce.cuh:63,148 and
detail/coalesced_reduction-inl.cuh (thin/medium/thick kernel regimes with
Kahan-compensated adds), detail/strided_reduction.cuh.

MI355X design: on GPU the contiguous-dim ("coalesced") reduction runs the
native wave64 kernel family from csrc/reductions.hip — logical-warp sizes
{2..64} chosen by row length, one-block-per-row for medium D, and a two-pass
grid for very long rows, with Kahan compensation for add-reductions (the
reference's numerics bar at D≥2^17). The strided (cross-row) reduction is a
column-parallel grid-stride kernel with in-LDS tree + device atomics. CPU path
is the torch oracle.
"""
from __future__ import annotations

from enum import Enum
from typing import Callable, Optional

import torch

from raft_amd._ext import require_ext
from raft_amd.utils import on_gpu


class Apply(Enum):
    ALONG_ROWS = 0      # reduce each row -> one value per row
    ALONG_COLUMNS = 1   # reduce each column -> one value per column


_MAIN_OPS = {
    "identity": lambda x: x,
    "sq": lambda x: x * x,
    "abs": lambda x: x.abs(),
    "sqrt": lambda x: x.abs().sqrt(),
}

_FINAL_OPS = {
    "identity": lambda x: x,
    "sqrt": lambda x: x.sqrt(),
}

# ops with a native GPU kernel (csrc/reductions.hip): (main_op, reduce_op)
_EXT_CODES = {("identity", "sum"): 0, ("sq", "sum"): 1, ("abs", "sum"): 2,
              ("identity", "max"): 3, ("identity", "min"): 4, ("abs", "max"): 5}


def coalesced_reduction(x: torch.Tensor, main_op: str = "identity",
                        reduce_op: str = "sum", final_op: str = "identity",
                        init: float = 0.0) -> torch.Tensor:
    """Reduce along the contiguous (last) dimension of a row-major matrix."""
    assert x.dim() == 2, "coalesced_reduction expects a 2D tensor"
    if on_gpu(x) and (main_op, reduce_op) in _EXT_CODES and x.dtype in (torch.float32, torch.float64):
        ext = require_ext()
        out = ext.reduce_rows(x.contiguous(), _EXT_CODES[(main_op, reduce_op)])
        return _FINAL_OPS[final_op](out)
    return _torch_reduce(x, dim=1, main_op=main_op, reduce_op=reduce_op, final_op=final_op)


def strided_reduction(x: torch.Tensor, main_op: str = "identity",
                      reduce_op: str = "sum", final_op: str = "identity") -> torch.Tensor:
    """Reduce along the strided (first) dimension of a row-major matrix."""
    assert x.dim() == 2
    if on_gpu(x) and (main_op, reduce_op) in _EXT_CODES and x.dtype in (torch.float32, torch.float64):
        ext = require_ext()
        out = ext.reduce_cols(x.contiguous(), _EXT_CODES[(main_op, reduce_op)])
        return _FINAL_OPS[final_op](out)
    return _torch_reduce(x, dim=0, main_op=main_op, reduce_op=reduce_op, final_op=final_op)


def _torch_reduce(x, dim, main_op, reduce_op, final_op):
    v = _MAIN_OPS[main_op](x.double() if x.dtype == torch.float32 else x)
    if reduce_op == "sum":
        r = v.sum(dim=dim)
    elif reduce_op == "max":
        r = v.max(dim=dim).values
    elif reduce_op == "min":
        r = v.min(dim=dim).values
    elif reduce_op == "prod":
        r = v.prod(dim=dim)
    else:
        raise ValueError(f"unknown reduce_op {reduce_op}")
    return _FINAL_OPS[final_op](r).to(x.dtype)


def reduce(x: torch.Tensor, apply: Apply = Apply.ALONG_ROWS, row_major: bool = True,
           main_op: str = "identity", reduce_op: str = "sum",
           final_op: str = "identity") -> torch.Tensor:
    """Unified dispatcher (reduce.cuh:148): picks coalesced vs strided by
    layout x direction, exactly as detail/reduce.cuh does."""
    if not row_major:
        x = x.t()
        apply = Apply.ALONG_COLUMNS if apply == Apply.ALONG_ROWS else Apply.ALONG_ROWS
    if apply == Apply.ALONG_ROWS:
        return coalesced_reduction(x, main_op, reduce_op, final_op)
    return strided_reduction(x, main_op, reduce_op, final_op)
