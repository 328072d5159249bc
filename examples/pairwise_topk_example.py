"""Pairwise distances + top-k selection composition (retrieval-style).

    python examples/pairwise_topk_example.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd.distance import pairwise_distance
from raft_amd.matrix import select_k


def main():
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    n_docs = 200_000 if dev == "cuda" else 5_000
    torch.manual_seed(0)
    docs = torch.randn(n_docs, 128, device=dev)
    queries = torch.randn(64, 128, device=dev)
    t0 = time.perf_counter()
    d = pairwise_distance(queries, docs, metric="sqeuclidean")
    vals, idx = select_k(d, 10, select_min=True)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    # verify against torch
    ref = torch.topk(torch.cdist(queries, docs) ** 2, 10, dim=1, largest=False)
    agree = float((idx == ref.indices).float().mean())
    print(f"device={dev} {64}x{n_docs}: top-10 in {dt*1e3:.1f} ms, "
          f"agreement vs torch {agree:.3f}")


if __name__ == "__main__":
    main()
