"""Composable operator functors (reference: cpp/include/raft/core/operators.hpp:27-391).

The reference ships a library of tiny device functors that public primitives
take as template parameters (reduce's main_op/final_op, map's f, ...). Here the
same composition surface is Python callables operating on torch tensors or
scalars — they vectorize through torch's fused elementwise kernels, and the
HIP kernels expose the common (op, reduce) pairs by enum code
(raft_amd/linalg/reduce.py _EXT_CODES) so the hot paths never cross Python.
"""
from __future__ import annotations

import torch


# ---- unary ----------------------------------------------------------------

def identity_op(x, *args):
    return x


def cast_op(dtype):
    def _op(x, *args):
        return x.to(dtype) if torch.is_tensor(x) else dtype(x)
    return _op


def key_op(kvp, *args):
    return kvp.key


def value_op(kvp, *args):
    return kvp.value


def sqrt_op(x, *args):
    return torch.sqrt(x) if torch.is_tensor(x) else x ** 0.5


def nz_op(x, *args):
    if torch.is_tensor(x):
        return (x != 0).to(x.dtype)
    return type(x)(x != 0)


def abs_op(x, *args):
    return torch.abs(x) if torch.is_tensor(x) else abs(x)


def sq_op(x, *args):
    return x * x


# ---- binary ---------------------------------------------------------------

def add_op(a, b):
    return a + b


def sub_op(a, b):
    return a - b


def mul_op(a, b):
    return a * b


def div_op(a, b):
    return a / b


def div_checkzero_op(a, b):
    if torch.is_tensor(a) or torch.is_tensor(b):
        a_t = a if torch.is_tensor(a) else torch.full_like(b, a)
        out = a_t / b
        return torch.where(b == 0, torch.zeros_like(out), out)
    return 0.0 if b == 0 else a / b


def pow_op(a, b):
    return a ** b


def mod_op(a, b):
    return a % b


def min_op(a, b):
    if torch.is_tensor(a) or torch.is_tensor(b):
        return torch.minimum(torch.as_tensor(a), torch.as_tensor(b))
    return min(a, b)


def max_op(a, b):
    if torch.is_tensor(a) or torch.is_tensor(b):
        return torch.maximum(torch.as_tensor(a), torch.as_tensor(b))
    return max(a, b)


def equal_op(a, b):
    return a == b


def notequal_op(a, b):
    return a != b


def greater_op(a, b):
    return a > b


def less_op(a, b):
    return a < b


def greater_or_equal_op(a, b):
    return a >= b


def less_or_equal_op(a, b):
    return a <= b


def argmin_op(a, b):
    """KVP reduce: keep the pair with the smaller value (ties -> smaller key)."""
    if b.value < a.value or (b.value == a.value and b.key < a.key):
        return b
    return a


def argmax_op(a, b):
    if b.value > a.value or (b.value == a.value and b.key < a.key):
        return b
    return a


# ---- composition ----------------------------------------------------------

def const_op(c):
    """Always return c (reference const_op)."""
    def _op(*args):
        return c
    return _op


def plug_const_op(c, op):
    """Bind the second argument of a binary op: x -> op(x, c)."""
    def _op(x, *args):
        return op(x, c)
    return _op


def add_const_op(c):
    return plug_const_op(c, add_op)


def sub_const_op(c):
    return plug_const_op(c, sub_op)


def mul_const_op(c):
    return plug_const_op(c, mul_op)


def div_const_op(c):
    return plug_const_op(c, div_op)


def pow_const_op(c):
    return plug_const_op(c, pow_op)


def mod_const_op(c):
    return plug_const_op(c, mod_op)


def equal_const_op(c):
    return plug_const_op(c, equal_op)


def compose_op(*ops):
    """compose_op(f, g, h)(x) = f(g(h(x))) — innermost applied first,
    matching the reference's compose_op nesting order."""
    def _op(x, *args):
        out = x
        for op in reversed(ops):
            out = op(out, *args)
        return out
    return _op


def map_args_op(op, *getters):
    """map_args_op(op, g1, g2)(args...) = op(g1(args...), g2(args...))."""
    def _op(*args):
        return op(*(g(*args) for g in getters))
    return _op


sqdiff_op = map_args_op(sq_op, sub_op)
absdiff_op = map_args_op(abs_op, sub_op)
