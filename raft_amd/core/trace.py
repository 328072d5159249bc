"""Tracing ranges (reference parity: raft/core/nvtx.hpp domain-scoped RAII
ranges behind RAFT_NVTX).

On ROCm, torch.cuda.nvtx lowers to rocTX markers, which rocprofv3 picks up
(--marker-trace). Ranges are compiled out unless RAFT_AMD_TRACE=1, mirroring
the reference's opt-in build flag.
"""
from __future__ import annotations

import functools
import os
from contextlib import contextmanager

import torch

_ENABLED = os.environ.get("RAFT_AMD_TRACE", "0") == "1" and torch.cuda.is_available()


@contextmanager
def annotate(name: str):
    """RAII-style range (common::nvtx::range parity)."""
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def annotated(name: str | None = None):
    """Decorator form: @annotated() wraps a primitive in a named range."""
    def deco(fn):
        rng_name = name or f"raft_amd::{fn.__module__.split('.')[-1]}::{fn.__name__}"

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            with annotate(rng_name):
                return fn(*args, **kwargs)

        return wrapper

    return deco
