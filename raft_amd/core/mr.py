"""Device memory resources — the MI355X-native RMM-equivalent layer.

Reference parity: rmm-backed resource accessors raft/core/resource/
resource_types.hpp:37-40 (WORKSPACE_RESOURCE / LARGE_WORKSPACE_RESOURCE),
mr/*.hpp adaptors (limiting, tracking/statistics), and the pool semantics the
reference inherits from rmm::mr::pool_memory_resource.

MI355X design: the base allocator is torch's caching allocator over the
288 GB HBM3E (already a pool — hipMalloc is never on the hot path), so the
value added here is the RMM *semantics* the reference's primitives assume:

  * PoolMemoryResource — slab sub-allocator with an enforced maximum size
    (workspace limit), 256 B alignment, first-fit + free-coalescing. Used for
    bounded workspaces: exceeding the cap raises MemoryLimitExceeded instead
    of OOMing the device.
  * LimitingAdaptor — cap any upstream by outstanding bytes.
  * TrackingAdaptor — allocation counts/bytes/peak + leak detection
    (memory_tracking_resources.hpp / memory_stats_resources.hpp parity).

Resources (core/resources.py) exposes a workspace resource slot wired to
these; chunked algorithms (kNN, pairwise) size their tiles from
`workspace_budget()` instead of hardcoded chunk rows.

Everything works on CPU tensors too (torch.empty on "cpu"), so the full
semantics are unit-tested without a GPU.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

ALIGN = 256


class MemoryLimitExceeded(RuntimeError):
    """Raised when an allocation would exceed the resource's byte limit."""


def _nbytes(shape, dtype) -> int:
    n = 1
    for s in (shape if isinstance(shape, (tuple, list)) else (shape,)):
        n *= int(s)
    return n * torch._utils._element_size(dtype)


class WorkspaceBuffer:
    """RAII-ish handle over a workspace allocation (device_uvector analog).

    Use as a context manager (preferred) or call .free() explicitly; the
    finalizer returns the block to the pool if the user forgets.
    """

    def __init__(self, mr: "DeviceMemoryResource", tensor: torch.Tensor, token):
        self._mr = mr
        self.tensor = tensor
        self._token = token
        self._freed = False

    def view(self, shape, dtype=torch.uint8) -> torch.Tensor:
        return self.tensor[: _nbytes(shape, dtype)].view(dtype).view(shape)

    def free(self):
        if not self._freed:
            self._freed = True
            self._mr._release(self._token)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.free()
        return False

    def __del__(self):  # safety net; explicit free/with is the contract
        try:
            self.free()
        except Exception:
            pass


class DeviceMemoryResource:
    """Abstract memory resource: allocate returns a WorkspaceBuffer of raw
    uint8 bytes; .view() reinterprets to the requested shape/dtype."""

    def allocate(self, nbytes: int) -> WorkspaceBuffer:
        raise NotImplementedError

    def allocate_tensor(self, shape, dtype=torch.float32) -> WorkspaceBuffer:
        return self.allocate(_nbytes(shape, dtype))

    def _release(self, token) -> None:
        raise NotImplementedError

    # -- stats (overridden by TrackingAdaptor/pool) -------------------------
    def outstanding_bytes(self) -> int:
        return 0

    def capacity_bytes(self) -> Optional[int]:
        """Byte cap, or None if unbounded."""
        return None

    def available_bytes(self) -> Optional[int]:
        cap = self.capacity_bytes()
        return None if cap is None else max(0, cap - self.outstanding_bytes())


class TorchMemoryResource(DeviceMemoryResource):
    """Upstream resource: torch caching allocator (the HBM3E pool)."""

    def __init__(self, device=None):
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
        self._lock = threading.Lock()
        self._outstanding = 0

    def allocate(self, nbytes: int) -> WorkspaceBuffer:
        t = torch.empty(int(nbytes), dtype=torch.uint8, device=self.device)
        with self._lock:
            self._outstanding += int(nbytes)
        return WorkspaceBuffer(self, t, int(nbytes))

    def _release(self, token) -> None:
        with self._lock:
            self._outstanding -= token

    def outstanding_bytes(self) -> int:
        return self._outstanding


class PoolMemoryResource(DeviceMemoryResource):
    """Slab sub-allocator with a hard maximum (rmm::pool_memory_resource
    + limiting semantics). First-fit over a sorted free list with
    coalescing on free; slabs grow geometrically up to maximum_pool_size."""

    def __init__(self, initial_pool_size: int = 1 << 20,
                 maximum_pool_size: Optional[int] = None, device=None):
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
        self.maximum_pool_size = maximum_pool_size
        self._lock = threading.Lock()
        self._slabs: List[torch.Tensor] = []
        # per-slab sorted free list of (offset, size)
        self._free: List[List[Tuple[int, int]]] = []
        self._outstanding = 0
        self._pool_bytes = 0
        self._next_slab = max(int(initial_pool_size), ALIGN)
        if maximum_pool_size is not None:
            self._next_slab = min(self._next_slab, int(maximum_pool_size))
        if self._next_slab:
            self._grow(self._next_slab)

    def _grow(self, nbytes: int) -> None:
        nbytes = (int(nbytes) + ALIGN - 1) // ALIGN * ALIGN
        if self.maximum_pool_size is not None and \
                self._pool_bytes + nbytes > self.maximum_pool_size:
            nbytes = self.maximum_pool_size - self._pool_bytes
            if nbytes <= 0:
                raise MemoryLimitExceeded(
                    f"pool at maximum {self.maximum_pool_size} bytes")
        self._slabs.append(torch.empty(nbytes, dtype=torch.uint8,
                                       device=self.device))
        self._free.append([(0, nbytes)])
        self._pool_bytes += nbytes

    def allocate(self, nbytes: int) -> WorkspaceBuffer:
        req = max((int(nbytes) + ALIGN - 1) // ALIGN * ALIGN, ALIGN)
        with self._lock:
            token = self._try_alloc(req)
            if token is None:
                # grow: geometric, at least req
                want = max(req, 2 * self._pool_bytes if self._pool_bytes else req)
                if self.maximum_pool_size is not None:
                    want = min(want, self.maximum_pool_size - self._pool_bytes)
                if want < req:
                    raise MemoryLimitExceeded(
                        f"allocation of {nbytes} bytes exceeds pool maximum "
                        f"{self.maximum_pool_size} (outstanding "
                        f"{self._outstanding}, pool {self._pool_bytes})")
                self._grow(want)
                token = self._try_alloc(req)
                if token is None:
                    raise MemoryLimitExceeded(
                        f"allocation of {nbytes} bytes exceeds pool maximum "
                        f"{self.maximum_pool_size}")
            slab_i, off = token
            self._outstanding += req
        t = self._slabs[slab_i][off:off + int(nbytes)]
        return WorkspaceBuffer(self, t, (slab_i, off, req))

    def _try_alloc(self, req: int):
        for si, freelist in enumerate(self._free):
            for fi, (off, size) in enumerate(freelist):
                if size >= req:
                    if size == req:
                        freelist.pop(fi)
                    else:
                        freelist[fi] = (off + req, size - req)
                    return (si, off)
        return None

    def _release(self, token) -> None:
        slab_i, off, req = token
        with self._lock:
            self._outstanding -= req
            fl = self._free[slab_i]
            # insert sorted + coalesce neighbors
            import bisect
            i = bisect.bisect_left(fl, (off, 0))
            fl.insert(i, (off, req))
            # coalesce with next
            if i + 1 < len(fl) and fl[i][0] + fl[i][1] == fl[i + 1][0]:
                fl[i] = (fl[i][0], fl[i][1] + fl[i + 1][1])
                fl.pop(i + 1)
            # coalesce with prev
            if i > 0 and fl[i - 1][0] + fl[i - 1][1] == fl[i][0]:
                fl[i - 1] = (fl[i - 1][0], fl[i - 1][1] + fl[i][1])
                fl.pop(i)

    def outstanding_bytes(self) -> int:
        return self._outstanding

    def pool_bytes(self) -> int:
        return self._pool_bytes

    def capacity_bytes(self) -> Optional[int]:
        return self.maximum_pool_size


class LimitingAdaptor(DeviceMemoryResource):
    """Cap an upstream resource by outstanding bytes
    (rmm limiting_resource_adaptor parity)."""

    def __init__(self, upstream: DeviceMemoryResource, limit_bytes: int):
        self.upstream = upstream
        self.limit_bytes = int(limit_bytes)
        self._lock = threading.Lock()
        self._outstanding = 0

    def allocate(self, nbytes: int) -> WorkspaceBuffer:
        with self._lock:
            if self._outstanding + int(nbytes) > self.limit_bytes:
                raise MemoryLimitExceeded(
                    f"allocation of {nbytes} bytes exceeds limit "
                    f"{self.limit_bytes} (outstanding {self._outstanding})")
            self._outstanding += int(nbytes)
        try:
            inner = self.upstream.allocate(nbytes)
        except Exception:
            with self._lock:
                self._outstanding -= int(nbytes)
            raise
        return WorkspaceBuffer(self, inner.tensor, (inner, int(nbytes)))

    def _release(self, token) -> None:
        inner, nbytes = token
        inner.free()
        with self._lock:
            self._outstanding -= nbytes

    def outstanding_bytes(self) -> int:
        return self._outstanding

    def capacity_bytes(self) -> Optional[int]:
        return self.limit_bytes


@dataclass
class AllocationStats:
    """memory_stats_resources.hpp:75 parity."""
    alloc_count: int = 0
    free_count: int = 0
    allocated_bytes: int = 0        # cumulative
    outstanding_bytes: int = 0
    peak_bytes: int = 0


class TrackingAdaptor(DeviceMemoryResource):
    """Tally allocations + detect leaks (memory_tracking_resources parity)."""

    def __init__(self, upstream: DeviceMemoryResource):
        self.upstream = upstream
        self._lock = threading.Lock()
        self.stats = AllocationStats()

    def allocate(self, nbytes: int) -> WorkspaceBuffer:
        inner = self.upstream.allocate(nbytes)
        with self._lock:
            s = self.stats
            s.alloc_count += 1
            s.allocated_bytes += int(nbytes)
            s.outstanding_bytes += int(nbytes)
            s.peak_bytes = max(s.peak_bytes, s.outstanding_bytes)
        return WorkspaceBuffer(self, inner.tensor, (inner, int(nbytes)))

    def _release(self, token) -> None:
        inner, nbytes = token
        inner.free()
        with self._lock:
            self.stats.free_count += 1
            self.stats.outstanding_bytes -= nbytes

    def outstanding_bytes(self) -> int:
        return self.stats.outstanding_bytes

    def capacity_bytes(self) -> Optional[int]:
        return self.upstream.capacity_bytes()

    def assert_no_leaks(self) -> None:
        if self.stats.outstanding_bytes != 0:
            raise RuntimeError(
                f"workspace leak: {self.stats.outstanding_bytes} bytes "
                f"outstanding ({self.stats.alloc_count} allocs, "
                f"{self.stats.free_count} frees)")


def default_workspace_budget(device) -> int:
    """Free-HBM estimate used when no explicit workspace cap is set: half of
    the currently free device memory (the other half stays for the caller's
    own tensors), or 1 GiB on CPU hosts."""
    device = torch.device(device)
    if device.type == "cuda" and torch.cuda.is_available():
        free, _total = torch.cuda.mem_get_info(device)
        return free // 2
    return 1 << 30
