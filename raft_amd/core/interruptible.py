"""Cooperative cancellation of stream synchronization.

Reference parity: raft/core/interruptible.hpp:64 (per-thread token; sync
spin-polls the stream and throws `interrupted_exception` when another thread
calls cancel()) and pylibraft/common/interruptible.pyx (SIGINT integration).
"""
from __future__ import annotations

import threading
import time
from contextlib import contextmanager

import torch


class InterruptedException(RuntimeError):
    pass


class Interruptible:
    _tls = threading.local()
    _registry: dict[int, "Interruptible"] = {}
    _reg_lock = threading.Lock()

    def __init__(self):
        self._cancelled = threading.Event()

    @classmethod
    def get_token(cls, thread_id: int | None = None) -> "Interruptible":
        if thread_id is None:
            tok = getattr(cls._tls, "token", None)
            if tok is None:
                tok = cls._tls.token = Interruptible()
                with cls._reg_lock:
                    cls._registry[threading.get_ident()] = tok
            return tok
        with cls._reg_lock:
            if thread_id not in cls._registry:
                cls._registry[thread_id] = Interruptible()
            return cls._registry[thread_id]

    def cancel(self) -> None:
        self._cancelled.set()

    def check(self) -> None:
        if self._cancelled.is_set():
            self._cancelled.clear()
            raise InterruptedException("raft_amd operation interrupted")

    def synchronize(self, stream: "torch.cuda.Stream | None" = None, poll_s: float = 0.0005) -> None:
        """Interruptible stream sync: poll query() and yield, as the reference
        spin-polls cudaStreamQuery (interruptible.hpp:76-93)."""
        self.check()
        if not torch.cuda.is_available():
            return
        if stream is None:
            stream = torch.cuda.current_stream()
        while not stream.query():
            self.check()
            time.sleep(poll_s)
        self.check()


def synchronize(stream=None) -> None:
    Interruptible.get_token().synchronize(stream)


@contextmanager
def cancel_on_sigint():  # pragma: no cover - signal path, manual use
    import signal

    token = Interruptible.get_token()
    prev = signal.getsignal(signal.SIGINT)

    def handler(signum, frame):
        token.cancel()

    signal.signal(signal.SIGINT, handler)
    try:
        yield token
    finally:
        signal.signal(signal.SIGINT, prev)
