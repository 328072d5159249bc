#!/usr/bin/env python3
"""Compile-time profile of the native build (reference parity:
cpp/scripts/analyze_nvcc_log.py — hipcc instead of nvcc).

Times each translation unit's hipcc compile separately and prints the
slowest first. Use when a header change regresses build latency.

Run: python scripts/analyze_build_log.py
"""
from __future__ import annotations

import subprocess
import sys
import tempfile
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))


def main() -> int:
    import build_ext as bx

    flags = bx.compile_flags()
    sources = sorted(bx.CSRC.glob("*.hip")) + sorted(bx.CSRC.glob("*.cpp"))
    times = []
    with tempfile.TemporaryDirectory() as td:
        for src in sources:
            obj = Path(td) / (src.stem + ".o")
            cmd = ["hipcc", "-c", "-x", "hip", str(src), "-o", str(obj)] + flags
            t0 = time.perf_counter()
            r = subprocess.run(cmd, capture_output=True)
            dt = time.perf_counter() - t0
            if r.returncode != 0:
                print(f"FAILED {src.name}:\n{r.stdout.decode()}{r.stderr.decode()}")
                return 1
            times.append((src.name, dt))
    for name, dt in sorted(times, key=lambda t: -t[1]):
        print(f"{dt:8.1f}s  {name}")
    print(f"{sum(t for _, t in times):8.1f}s  TOTAL (serial)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
