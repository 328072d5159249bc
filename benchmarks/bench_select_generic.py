#!/usr/bin/env python3
"""Generic select-k engine (fp64 / bf16 / large-k) vs torch.topk.

The generic threshold+filter engine is coverage-first (NOTES round-3 item
4); this harness puts numbers on it so the trade is recorded.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd.matrix import select_k


def bench(name, fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t) / iters * 1e3
    print(f"{name:44s} {ms:8.3f} ms")
    return ms


def main():
    torch.manual_seed(0)
    cases = [
        ("fp64 [4096 x 100k] k=64", torch.float64, 4096, 100_000, 64),
        ("fp32 [4096 x 100k] k=4096 (large-k)", torch.float32, 4096, 100_000, 4096),
        ("bf16 [8192 x 100k] k=64", torch.bfloat16, 8192, 100_000, 64),
    ]
    for name, dt, b, n, k in cases:
        x = torch.randn(b, n, device="cuda").to(dt)
        r1 = bench(f"select_k {name}", lambda: select_k(x, k))
        r2 = bench(f"torch.topk {name}", lambda: torch.topk(x, k, largest=False))
        print(f"  ratio select_k/topk = {r1 / r2:.2f}x")
        v, i = select_k(x, k)
        rv, ri = torch.topk(x, k, largest=False)
        torch.testing.assert_close(v.float().sort(dim=1).values,
                                   rv.float().sort(dim=1).values,
                                   atol=1e-3, rtol=1e-3)
    print("OK")


if __name__ == "__main__":
    main()
