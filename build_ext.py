#!/usr/bin/env python3
"""In-tree build of the raft_amd native extension (raft_amd/_C*.so).

Drives hipcc directly (no JIT cache: the built .so must live in-tree so it
travels to GPU boxes with the repo snapshot). Cross-compiles for gfx950 —
works on GPU-less CI hosts.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "csrc"
BUILD = ROOT / "build" / "ext"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_flags():
    import torch
    import torch.utils.cpp_extension as ce

    inc = ce.include_paths(device_type="cuda")
    lib = ce.library_paths(device_type="cuda")
    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    return inc, lib, abi


def compile_flags(inc=None, abi=None):
    """The per-TU hipcc flag set (shared with scripts/analyze_build_log.py)."""
    if inc is None or abi is None:
        inc, _, abi = _torch_flags()
    py_inc = sysconfig.get_paths()["include"]
    return [
        "-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_C", "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-Wno-unused-result", "-Wno-ignored-attributes",
    ] + [f"-I{p}" for p in inc + [py_inc, str(CSRC)]]


def build(verbose: bool = True) -> Path:
    inc, libdirs, abi = _torch_flags()
    common = compile_flags(inc, abi)
    BUILD.mkdir(parents=True, exist_ok=True)

    sources = sorted(CSRC.glob("*.hip")) + sorted(CSRC.glob("*.cpp"))
    headers_mtime = max(h.stat().st_mtime for h in CSRC.glob("*.h"))

    objs = []
    procs = []
    for src in sources:
        obj = BUILD / (src.stem + ".o")
        objs.append(obj)
        if obj.exists() and obj.stat().st_mtime > max(src.stat().st_mtime,
                                                      headers_mtime):
            continue
        cmd = ["hipcc", "-c", "-x", "hip", str(src), "-o", str(obj)] + common
        if verbose:
            print("[build_ext] compile", src.name, flush=True)
        procs.append((src, subprocess.Popen(cmd, stdout=subprocess.PIPE,
                                            stderr=subprocess.STDOUT)))
    failed = False
    for src, p in procs:
        out, _ = p.communicate()
        if p.returncode != 0:
            failed = True
            print(f"[build_ext] FAILED {src.name}:\n{out.decode()}", flush=True)
        elif out.strip() and verbose:
            print(out.decode(), flush=True)
    if failed:
        raise RuntimeError("hipcc compilation failed")

    ext_suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    out_so = ROOT / "raft_amd" / f"_C{ext_suffix}"
    link = (["hipcc", "-shared", "-fPIC", "-o", str(out_so)] + [str(o) for o in objs]
            + [f"-L{d}" for d in libdirs]
            + ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
               "-ltorch_python", "-lrocblas", "-lhipblaslt", "-lamdhip64"]
            + [f"-Wl,-rpath,{d}" for d in libdirs])
    if verbose:
        print("[build_ext] link", out_so.name, flush=True)
    r = subprocess.run(link, capture_output=True)
    if r.returncode != 0:
        raise RuntimeError(f"link failed:\n{r.stdout.decode()}\n{r.stderr.decode()}")
    print(f"[build_ext] built {out_so}", flush=True)
    return out_so


if __name__ == "__main__":
    build()
