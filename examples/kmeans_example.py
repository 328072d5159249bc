"""k-means on synthetic blobs — the flagship workflow.

Runs on CPU (torch oracle path) or any MI355X (native HIP engines).
    python examples/kmeans_example.py [--rows 100000] [--k 64]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd.cluster import KMeans
from raft_amd.random import make_blobs, RngState


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=None,
                    help="default: 1M on GPU, 20k on CPU (oracle path is slow)")
    ap.add_argument("--dim", type=int, default=64)
    ap.add_argument("--k", type=int, default=64)
    args = ap.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    if args.rows is None:
        args.rows = 1_000_000 if dev == "cuda" else 20_000
    x, y_true, centers = make_blobs(args.rows, args.dim, n_clusters=args.k,
                                    cluster_std=0.5, state=RngState(seed=0),
                                    device=dev)
    km = KMeans(n_clusters=args.k, max_iter=25, init="scalable", n_init=1)
    t0 = time.perf_counter()
    labels = km.fit_predict(x)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    d = torch.cdist(centers, km.cluster_centers_)
    recovered = int((d.min(dim=1).values < 2.0).sum())
    print(f"device={dev} rows={args.rows} k={args.k}: fit in {dt:.3f}s, "
          f"inertia={km.inertia_:.4g}, {recovered}/{args.k} true centers recovered")


if __name__ == "__main__":
    main()
