"""Row gather/scatter (reference: raft/matrix/gather.cuh, scatter_inplace).

torch.index_select/index_copy are the ROCm vendor kernels for these
memory-bound patterns; gather_if adds the transform/conditional variants.
"""
from __future__ import annotations

from typing import Callable, Optional

import torch


def gather(x: torch.Tensor, indices: torch.Tensor, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Gather rows x[indices[i]] (reference matrix::gather)."""
    res = torch.index_select(x, 0, indices.to(torch.int64))
    if out is not None:
        out.copy_(res)
        return out
    return res


def gather_if(x: torch.Tensor, indices: torch.Tensor, stencil: torch.Tensor,
              pred: Callable[[torch.Tensor], torch.Tensor],
              transform: Optional[Callable] = None) -> torch.Tensor:
    """Gather rows x[indices[i]] where pred(stencil[i]); others keep zeros."""
    rows = gather(x, indices)
    if transform is not None:
        rows = transform(rows)
    mask = pred(stencil).to(torch.bool)
    out = torch.zeros_like(rows)
    out[mask] = rows[mask]
    return out


def scatter(x: torch.Tensor, indices: torch.Tensor, src: torch.Tensor | None = None) -> torch.Tensor:
    """If src given: x[indices[i], :] = src[i, :]. Else permute-in-place
    semantics: out[indices[i], :] = x[i, :] (reference scatter_inplace)."""
    if src is not None:
        x.index_copy_(0, indices.to(torch.int64), src)
        return x
    out = torch.empty_like(x)
    out.index_copy_(0, indices.to(torch.int64), x)
    return out


def gather_inplace(x: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
    """x[i, :] = x[indices[i], :] (reference gather_inplace; the reference
    uses cycle-following to avoid the copy — HBM3E bandwidth makes the
    double-buffered form faster here)."""
    x.copy_(torch.index_select(x, 0, indices.to(torch.int64)))
    return x


def scatter_inplace(x: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
    """x[indices[i], :] = x[i, :] in place (reference scatter_inplace)."""
    out = torch.empty_like(x)
    out.index_copy_(0, indices.to(torch.int64), x)
    x.copy_(out)
    return x
