"""mdbuffer — location-polymorphic buffer.

Reference parity: raft/core/mdbuffer.cuh (std::variant over host/device/
managed/pinned owning/view alternatives with lazy copy-on-view-request) and
memory_type_dispatcher.cuh (route to kernels by location).

Here the variant is a per-location torch tensor cache: `view(memory_type)`
returns (copying lazily, caching) the tensor in the requested location.
"""
from __future__ import annotations

from enum import Enum
from typing import Dict

import torch


class MemoryType(Enum):
    HOST = "host"
    DEVICE = "device"
    PINNED = "pinned"
    MANAGED = "managed"   # no HIP managed allocs through torch: alias of device


class MDBuffer:
    def __init__(self, data: torch.Tensor):
        self._views: Dict[MemoryType, torch.Tensor] = {}
        mt = MemoryType.DEVICE if data.is_cuda else (
            MemoryType.PINNED if data.is_pinned() else MemoryType.HOST)
        self._views[mt] = data
        self._origin = mt

    @property
    def memory_type(self) -> MemoryType:
        return self._origin

    def view(self, memory_type: MemoryType = None) -> torch.Tensor:
        """Tensor in the requested location; lazily copied and cached."""
        if memory_type is None:
            memory_type = self._origin
        if memory_type == MemoryType.MANAGED:
            memory_type = MemoryType.DEVICE
        if memory_type in self._views:
            return self._views[memory_type]
        src = self._views[self._origin]
        if memory_type == MemoryType.DEVICE:
            out = src.cuda()
        elif memory_type == MemoryType.PINNED:
            out = src.cpu().pin_memory() if torch.cuda.is_available() else src.cpu()
        else:
            out = src.cpu()
        self._views[memory_type] = out
        return out

    def is_cached(self, memory_type: MemoryType) -> bool:
        return memory_type in self._views


def memory_type_dispatcher(buf: "MDBuffer | torch.Tensor", device_fn, host_fn):
    """Route by location (memory_type_dispatcher.cuh parity)."""
    t = buf.view() if isinstance(buf, MDBuffer) else buf
    return device_fn(t) if t.is_cuda else host_fn(t)


def copy_mdspan(dst: torch.Tensor, src: torch.Tensor) -> torch.Tensor:
    """Layout/location/dtype-converting copy (reference: core/detail/copy.hpp
    mdspan_copyable dispatch — cudaMemcpy for same-layout, tiled conversion
    kernel otherwise). Torch's copy_ engine performs the same dispatch on
    ROCm: a contiguous same-dtype pair becomes one hipMemcpy; transposed or
    dtype-converting pairs run a vectorized conversion kernel; cross-device
    pairs stage through DMA. Shapes must match; dst's layout/device/dtype
    win."""
    if dst.shape != src.shape:
        raise ValueError(f"shape mismatch {tuple(dst.shape)} vs {tuple(src.shape)}")
    dst.copy_(src)
    return dst
