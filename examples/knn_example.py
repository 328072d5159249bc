"""Brute-force exact kNN over a large bf16 index (sample->filter engine on GPU).

    python examples/knn_example.py [--index-rows 1000000] [--queries 1000]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd.neighbors import knn


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--index-rows", type=int, default=200_000)
    ap.add_argument("--dim", type=int, default=128)
    ap.add_argument("--queries", type=int, default=1000)
    ap.add_argument("--k", type=int, default=10)
    args = ap.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    x = torch.randn(args.index_rows, args.dim, device=dev)
    if dev == "cuda":
        x = x.bfloat16()          # bf16 index: native MFMA path
    q = x[: args.queries].clone()
    t0 = time.perf_counter()
    dists, idx = knn(x, q, k=args.k)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    self_hit = float((idx[:, 0] == torch.arange(args.queries, device=dev)).float().mean())
    print(f"device={dev} index={args.index_rows}x{args.dim} q={args.queries} "
          f"k={args.k}: {dt:.3f}s ({args.queries/dt:.0f} q/s), self-hit {self_hit:.3f}")


if __name__ == "__main__":
    main()
