"""host_span / device_span (reference: core/span.hpp, device_span.hpp).

Torch tensors already carry (pointer, length, device) — a span here is a
validated non-owning 1-D view with the reference's location contract:
device_span asserts device residency, host_span asserts host residency.
Subspans are zero-copy slices of the same storage.
"""
from __future__ import annotations

import torch


def _as_span(t: torch.Tensor, want_cuda: bool, kind: str) -> torch.Tensor:
    if not t.is_contiguous():
        # reshape of a strided tensor would COPY — that breaks the
        # non-owning contract; spans only exist over contiguous storage
        raise TypeError(f"{kind} requires contiguous storage")
    v = t.view(-1)
    if v.is_cuda != want_cuda:
        raise TypeError(f"{kind} requires a {'device' if want_cuda else 'host'}"
                        f" tensor, got device={t.device}")
    return v


def host_span(t: torch.Tensor) -> torch.Tensor:
    """Non-owning flat view over host memory."""
    return _as_span(t, want_cuda=False, kind="host_span")


def device_span(t: torch.Tensor) -> torch.Tensor:
    """Non-owning flat view over device (HBM3E) memory."""
    return _as_span(t, want_cuda=True, kind="device_span")


def subspan(span: torch.Tensor, offset: int, count: int | None = None) -> torch.Tensor:
    """span.subspan(offset, count) — zero-copy, same storage."""
    end = span.numel() if count is None else offset + count
    if offset < 0 or end > span.numel():
        raise IndexError(f"subspan [{offset}, {end}) out of range {span.numel()}")
    return span[offset:end]
