#!/usr/bin/env python3
"""Header/source hygiene checker (reference parity: cpp/scripts/include_checker.py).

Checks, for csrc/:
  * every .h header has #pragma once
  * kernel translation units (.hip) do not include torch headers (keeps
    device code re-compilable without the torch toolchain; torch types stay
    in bindings.cpp)
  * no file includes a quoted path that does not exist next to it
  * no CUDA compatibility includes (cuda_runtime.h etc.) — this tree is
    HIP/CDNA4-only by design

Exit code 1 on any violation. Run: python scripts/include_checker.py
"""
from __future__ import annotations

import re
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
CSRC = ROOT / "csrc"

FORBIDDEN_IN_HIP = ("torch/", "ATen/", "pybind11/")
FORBIDDEN_ANYWHERE = ("cuda_runtime", "cublas", "cusparse", "cusolver",
                      "curand", "nccl.h", "cub/", "thrust/")


def main() -> int:
    bad = 0
    files = sorted(CSRC.glob("*")) + sorted((ROOT / "include" / "raft_amd").glob("*.hpp"))
    for f in files:
        if f.suffix not in (".h", ".hpp", ".hip", ".cpp"):
            continue
        text = f.read_text()
        rel = f.relative_to(ROOT)
        if f.suffix == ".h" and "#pragma once" not in text:
            print(f"{rel}: header missing '#pragma once'")
            bad += 1
        for m in re.finditer(r'#include\s+"([^"]+)"', text):
            inc = m.group(1)
            if not (f.parent / inc).exists():
                print(f"{rel}: quoted include not found: {inc}")
                bad += 1
        for m in re.finditer(r"#include\s+<([^>]+)>", text):
            inc = m.group(1)
            if any(inc.startswith(p) for p in FORBIDDEN_ANYWHERE):
                print(f"{rel}: forbidden CUDA-ecosystem include <{inc}>")
                bad += 1
            if f.suffix == ".hip" and any(inc.startswith(p) for p in FORBIDDEN_IN_HIP):
                print(f"{rel}: torch include <{inc}> in a kernel TU "
                      f"(keep torch types in bindings.cpp)")
                bad += 1
    if bad:
        print(f"include_checker: {bad} violation(s)")
        return 1
    print("include_checker: OK")
    return 0


if __name__ == "__main__":
    sys.exit(main())
