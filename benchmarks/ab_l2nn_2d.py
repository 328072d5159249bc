"""A/B: v1 col-tile-loop kernel vs 2D XCD-swizzled tile-pair kernel."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from raft_amd.neighbors.fused_l2nn import fused_l2nn_presplit, split_bf16_slices

def run(m=10_000_000, n=1024, d=256, iters=5):
    torch.manual_seed(0)
    x = torch.randn(m, d, device="cuda")
    c = torch.randn(n, d, device="cuda")
    xs = split_bf16_slices(x, 2); xn = (x * x).sum(1)
    # correctness first (small slice covers edge rows via m%128!=0)
    xs_s = [s[:1200001].contiguous() for s in xs]
    import raft_amd._C as C
    cs = split_bf16_slices(c, 2); cn = (c * c).sum(1)

    d1 = C.fused_l2nn_split(xs_s, cs, xn[:1200001], cn)
    print("mode:", os.environ.get("_MODE"), "sample dmin/amin hash:",
          float(d1[0].double().sum()), int(d1[1].long().sum()), float(d1[2].double().sum()))
    # timing at full m
    for _ in range(2):
        C.fused_l2nn_split(xs, cs, xn, cn)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        C.fused_l2nn_split(xs, cs, xn, cn)
    torch.cuda.synchronize()
    print(f"kernel path avg: {(time.perf_counter()-t0)/iters*1000:.3f} ms")

run()
