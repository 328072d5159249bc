"""Elementwise map machinery.

Reference parity: raft/linalg/map.cuh:95-173 (variadic vectorized map kernel),
map_then_reduce.cuh, and the thin arithmetic wrappers
(add/subtract/multiply/divide/power/sqrt/eltwise, binary/unary/ternary_op).

MI355X note: torch elementwise ops on ROCm are already vectorized HIP kernels
saturating HBM3E; a bespoke map kernel would duplicate them with no headroom
(these ops are memory-bound). The framework's fusion story on GPU is instead
*kernel-level* fusion inside the hot primitives (distance epilogues, normalize,
make_blobs), where fusion actually saves HBM round-trips.
"""
from __future__ import annotations

from typing import Callable

import torch


def map_op(fn: Callable, *tensors: torch.Tensor) -> torch.Tensor:
    """out[i] = fn(t0[i], t1[i], ...) — the N-ary map (map.cuh:95)."""
    return fn(*tensors)


def map_offset(fn: Callable, n: int, *tensors: torch.Tensor, device=None,
               dtype=torch.float32) -> torch.Tensor:
    """out[i] = fn(i, t0[i], ...) — map with the flat offset as first arg."""
    if tensors:
        device = tensors[0].device
    idx = torch.arange(n, device=device)
    return fn(idx, *tensors)


def unary_op(fn, x):
    """Apply fn(x) elementwise (reference unary_op)."""
    return fn(x)


def binary_op(fn, x, y):
    """Apply fn(x, y) elementwise (reference binary_op)."""
    return fn(x, y)


def ternary_op(fn, x, y, z):
    """Apply fn(x, y, z) elementwise (reference ternary_op)."""
    return fn(x, y, z)


def add(x, y):
    """Elementwise a + b (map family)."""
    return x + y


def subtract(x, y):
    """Elementwise a - b (map family)."""
    return x - y


def multiply(x, y):
    """Elementwise a * b (map family)."""
    return x * y


def divide(x, y):
    """Elementwise a / b (map family)."""
    return x / y


def power(x, y):
    """Elementwise power (map family)."""
    return torch.pow(x, y)


def sqrt(x):
    """Elementwise sqrt (map family)."""
    return torch.sqrt(x)


def eltwise(fn, *tensors):
    """Variadic elementwise apply (reference eltwise)."""
    return fn(*tensors)


def map_then_reduce(fn: Callable, reduce_op: str, *tensors: torch.Tensor):
    """Fused elementwise + full reduction (map_then_reduce.cuh)."""
    v = fn(*tensors)
    if reduce_op == "sum":
        return v.sum()
    if reduce_op == "max":
        return v.max()
    if reduce_op == "min":
        return v.min()
    raise ValueError(f"unknown reduce_op {reduce_op}")


map_reduce = map_then_reduce
