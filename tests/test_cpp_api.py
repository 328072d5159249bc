"""Compile-and-run test of the public C++ API (include/raft_amd/raft_amd.hpp)
against the build/ext kernel objects — proves the native layer is consumable
from C++ with no Python (the raft_runtime consumability property)."""
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")
def test_cpp_smoke(tmp_path):
    objs = [os.path.join(ROOT, "build", "ext", f"{n}.o")
            for n in ("reductions", "rng")]
    for o in objs:
        if not os.path.exists(o):
            pytest.skip("build/ext objects not present (run build_ext.py)")
    exe = str(tmp_path / "cpp_smoke")
    obj = str(tmp_path / "smoke.o")
    # compile then link in two steps (hipcc treats .o inputs as sources when
    # mixed with .cpp under --offload-arch)
    r = subprocess.run(["hipcc", "--offload-arch=gfx950", "-O2", "-std=c++17",
                        "-c", os.path.join(ROOT, "tests", "cpp", "smoke.cpp"),
                        f"-I{os.path.join(ROOT, 'include')}", "-o", obj],
                       capture_output=True, timeout=300)
    assert r.returncode == 0, r.stderr.decode()
    r = subprocess.run(["hipcc", obj, *objs, "-o", exe],
                       capture_output=True, timeout=300)
    assert r.returncode == 0, r.stderr.decode()
    r = subprocess.run([exe], capture_output=True, timeout=120)
    assert r.returncode == 0, (r.stdout.decode(), r.stderr.decode())
    assert b"cpp smoke OK" in r.stdout
