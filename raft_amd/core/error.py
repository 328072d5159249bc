"""Exceptions + expects/fail helpers (reference: cpp/include/raft/core/error.hpp).

The reference's RAFT_EXPECTS/RAFT_FAIL attach a backtrace to raft::exception;
Python tracebacks carry that for free, so RaftError only needs the hierarchy
and the check helpers the rest of the package raises through.
"""
from __future__ import annotations


class RaftError(RuntimeError):
    """Base exception (reference raft::exception)."""


class LogicError(RaftError):
    """Precondition violation (reference raft::logic_error / RAFT_EXPECTS)."""


class HipError(RaftError):
    """Device-side failure surfaced from a HIP call (reference cuda_error)."""


def expects(cond: bool, msg: str = "precondition violated") -> None:
    """RAFT_EXPECTS: raise LogicError unless cond."""
    if not cond:
        raise LogicError(msg)


def fail(msg: str) -> None:
    """RAFT_FAIL: unconditional LogicError."""
    raise LogicError(msg)
