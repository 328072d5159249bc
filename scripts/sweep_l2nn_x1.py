"""Schedule sweep for the 1-slice fused L2-NN kernel (run on the GPU box).

Times ext.fused_l2nn_split alone (no verify pass) on 10M x 256, k=1024 —
the bf16x1v flagship composition — across the 2d engine's env-selected
schedule variants. Env toggles are cached per process, so each config runs
in a subprocess.
"""
import json
import os
import subprocess
import sys

WORKER = r"""
import time, torch
from raft_amd._ext import require_ext
ext = require_ext()
m, d, k = 10_000_000, 256, 1024
torch.manual_seed(0)
x = torch.randn(m, d, device="cuda")
c = torch.randn(k, d, device="cuda")
xs = [x.bfloat16().contiguous()]
cs = [c.bfloat16().contiguous()]
xn = (x * x).sum(dim=1)
cn = (c * c).sum(dim=1)
for _ in range(3):
    ext.fused_l2nn_split(xs, cs, xn, cn)
torch.cuda.synchronize()
t0 = time.perf_counter()
N = 15
for _ in range(N):
    ext.fused_l2nn_split(xs, cs, xn, cn)
torch.cuda.synchronize()
print(f"RESULT {(time.perf_counter() - t0) / N * 1e3:.3f} ms")
"""

CONFIGS = {
    "default(GT8)": {},
    "GT1": {"RAFT_AMD_L2NN_GT": "1"},
    "GT4": {"RAFT_AMD_L2NN_GT": "4"},
    "BK32": {"RAFT_AMD_L2NN_BK32": "1"},
    "BK32+GT4": {"RAFT_AMD_L2NN_BK32": "1", "RAFT_AMD_L2NN_GT": "4"},
    "AD": {"RAFT_AMD_L2NN_AD": "1"},
    "eng256": {"RAFT_AMD_L2NN_256": "1"},
}

if __name__ == "__main__":
    for name, env in CONFIGS.items():
        e = dict(os.environ, **env)
        r = subprocess.run([sys.executable, "-c", WORKER], env=e,
                           capture_output=True, text=True, timeout=300)
        line = [l for l in r.stdout.splitlines() if l.startswith("RESULT")]
        print(f"{name:14s} {line[0][7:] if line else 'FAIL ' + r.stderr[-200:]}",
              flush=True)
