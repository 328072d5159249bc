"""Logging for raft_amd (reference parity: raft/core/logger.hpp:25-49).

stderr by default; RAFT_AMD_DEBUG env var redirects to a file sink; level via
set_level or RAFT_AMD_LOG_LEVEL (trace/debug/info/warn/error/critical/off).
"""
from __future__ import annotations

import logging
import os
import sys

_LEVELS = {
    "trace": 5,
    "debug": logging.DEBUG,
    "info": logging.INFO,
    "warn": logging.WARNING,
    "error": logging.ERROR,
    "critical": logging.CRITICAL,
    "off": logging.CRITICAL + 10,
}

logging.addLevelName(5, "TRACE")
_logger: logging.Logger | None = None


def get_logger() -> logging.Logger:
    global _logger
    if _logger is None:
        _logger = logging.getLogger("raft_amd")
        debug_file = os.environ.get("RAFT_AMD_DEBUG")
        handler = logging.FileHandler(debug_file) if debug_file else logging.StreamHandler(sys.stderr)
        handler.setFormatter(logging.Formatter("[%(levelname)s] [%(asctime)s] %(message)s"))
        _logger.addHandler(handler)
        lvl = os.environ.get("RAFT_AMD_LOG_LEVEL", "warn").lower()
        _logger.setLevel(_LEVELS.get(lvl, logging.WARNING))
    return _logger


def set_level(level: str) -> None:
    get_logger().setLevel(_LEVELS[level.lower()])
