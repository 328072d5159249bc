"""Memory observability.

Reference parity: raft/core/memory_stats_resources.hpp (alloc counts/bytes),
memory_tracking_resources.hpp (tracking adaptors) and mr/resource_monitor.hpp
(background sampling thread streaming samples).

On MI355X the device pool is torch's caching allocator over the 288 GB HBM3E;
these helpers expose its counters in the reference's shapes.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Callable, List, Optional

import torch


@dataclass
class MemoryStats:
    """Allocation tally (memory_stats_resources parity)."""
    allocated_bytes: int = 0
    reserved_bytes: int = 0
    alloc_count: int = 0
    peak_allocated_bytes: int = 0

    @classmethod
    def capture(cls, device=None) -> "MemoryStats":
        if not torch.cuda.is_available():
            return cls()
        s = torch.cuda.memory_stats(device)
        return cls(
            allocated_bytes=s.get("allocated_bytes.all.current", 0),
            reserved_bytes=s.get("reserved_bytes.all.current", 0),
            alloc_count=s.get("allocation.all.allocated", 0),
            peak_allocated_bytes=s.get("allocated_bytes.all.peak", 0),
        )


class TrackingScope:
    """Context manager tallying allocations inside the scope
    (memory_tracking_resources adaptor parity)."""

    def __init__(self, device=None):
        self.device = device
        self.before: Optional[MemoryStats] = None
        self.after: Optional[MemoryStats] = None

    def __enter__(self):
        self.before = MemoryStats.capture(self.device)
        return self

    def __exit__(self, *exc):
        self.after = MemoryStats.capture(self.device)
        return False

    @property
    def delta_allocated(self) -> int:
        return self.after.allocated_bytes - self.before.allocated_bytes

    @property
    def delta_alloc_count(self) -> int:
        return self.after.alloc_count - self.before.alloc_count


class ResourceMonitor:
    """Background sampling thread (mr/resource_monitor.hpp:42 parity):
    samples allocator state at a fixed period and hands each sample to a sink
    callable (or collects them)."""

    def __init__(self, period_s: float = 0.05,
                 sink: Optional[Callable[[float, MemoryStats], None]] = None,
                 device=None):
        self.period_s = period_s
        self.sink = sink
        self.device = device
        self.samples: List[tuple] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self) -> "ResourceMonitor":
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()
        return self

    def _run(self):
        t0 = time.perf_counter()
        while not self._stop.is_set():
            s = MemoryStats.capture(self.device)
            t = time.perf_counter() - t0
            if self.sink:
                self.sink(t, s)
            else:
                self.samples.append((t, s))
            self._stop.wait(self.period_s)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
        return False


class TemporaryDeviceBuffer:
    """Workspace-backed scratch with host fallback
    (reference: core/temporary_device_buffer.hpp).

    Wraps `data` for device consumption: if `data` is already on the target
    device it is passed through (zero-copy unless write_back demands
    isolation); a host tensor is copied in, and copied back on exit when
    write_back=True.
    """

    def __init__(self, data: "torch.Tensor", device=None, write_back: bool = False):
        self._src = data
        self._write_back = write_back
        dev = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available() else data.device)
        # same device: pass through zero-copy (writes hit `data` directly);
        # otherwise copy in (and back on exit when write_back)
        self._buf = data if data.device == dev else data.to(dev)

    def view(self) -> "torch.Tensor":
        return self._buf

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        if self._write_back and self._buf.data_ptr() != self._src.data_ptr():
            self._src.copy_(self._buf.to(self._src.device))
        return False
