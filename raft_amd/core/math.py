"""Scalar/tensor-safe math wrappers (reference: cpp/include/raft/core/math.hpp).

The reference provides host/device-overloaded wrappers so templated code can
call raft::log/exp/... on any arithmetic type; here the same names accept
Python scalars or torch tensors uniformly (torch dispatches device-side).
"""
from __future__ import annotations

import math as _pymath

import torch


def _t(fn_t, fn_s, x):
    return fn_t(x) if torch.is_tensor(x) else fn_s(x)


def abs(x):  # noqa: A001 - mirrors reference name
    return _t(torch.abs, _pymath.fabs, x)


def exp(x):
    return _t(torch.exp, _pymath.exp, x)


def log(x):
    return _t(torch.log, _pymath.log, x)


def sqrt(x):
    """Scalar/tensor-safe sqrt (reference math.hpp)."""
    return _t(torch.sqrt, _pymath.sqrt, x)


def sin(x):
    return _t(torch.sin, _pymath.sin, x)


def cos(x):
    return _t(torch.cos, _pymath.cos, x)


def tanh(x):
    return _t(torch.tanh, _pymath.tanh, x)


def asin(x):
    return _t(torch.asin, _pymath.asin, x)


def acos(x):
    return _t(torch.acos, _pymath.acos, x)


def atan(x):
    return _t(torch.atan, _pymath.atan, x)


def atan2(y, x):
    if torch.is_tensor(y) or torch.is_tensor(x):
        return torch.atan2(torch.as_tensor(y), torch.as_tensor(x))
    return _pymath.atan2(y, x)


def pow(x, y):  # noqa: A001
    return x ** y


def sigmoid(x):
    """Numerically-stable logistic (reference math.hpp sigmoid)."""
    if torch.is_tensor(x):
        return torch.sigmoid(x)
    if x >= 0:
        z = _pymath.exp(-x)
        return 1.0 / (1.0 + z)
    z = _pymath.exp(x)
    return z / (1.0 + z)


def log1p(x):
    return _t(torch.log1p, _pymath.log1p, x)


def expm1(x):
    return _t(torch.expm1, _pymath.expm1, x)


def max(a, b):  # noqa: A001
    if torch.is_tensor(a) or torch.is_tensor(b):
        return torch.maximum(torch.as_tensor(a), torch.as_tensor(b))
    return a if a >= b else b


def min(a, b):  # noqa: A001
    if torch.is_tensor(a) or torch.is_tensor(b):
        return torch.minimum(torch.as_tensor(a), torch.as_tensor(b))
    return a if a <= b else b
