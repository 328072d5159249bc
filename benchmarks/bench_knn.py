#!/usr/bin/env python3
"""BASELINE config 5: fused L2-NN + k-select (k=64), bf16, 100M rows chunked.

Brute-force kNN of a query batch against a 100M x 128 bf16 index (25.6 GB,
HBM-resident — the 288 GB sizing case). Per (query-chunk, index-chunk) tile:
bf16 MFMA GEMM (rocBLAS) + fused L2 epilogue + native radix select_k, then a
k-way merge select over index chunks. Metric: query rows/sec (and the
distance throughput implied).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--index-rows", type=int, default=100_000_000)
    p.add_argument("--dim", type=int, default=128)
    p.add_argument("--queries", type=int, default=16384)
    p.add_argument("--k", type=int, default=64)
    p.add_argument("--query-chunk", type=int, default=8192)
    p.add_argument("--index-chunk", type=int, default=1_000_000)
    args = p.parse_args()

    assert torch.cuda.is_available()
    dev = torch.device("cuda")
    from raft_amd.random import RngState
    from raft_amd.random.rng import normal
    from raft_amd.neighbors.brute_force import knn

    # index generated in fp32 chunks then cast (peak mem: index + one chunk)
    t0 = time.perf_counter()
    x = torch.empty((args.index_rows, args.dim), dtype=torch.bfloat16, device=dev)
    state = RngState(seed=3)
    gchunk = 10_000_000
    for s in range(0, args.index_rows, gchunk):
        e = min(s + gchunk, args.index_rows)
        x[s:e] = normal(((e - s), args.dim), state=state, device=dev).bfloat16()
    q = x[: args.queries].clone()   # queries drawn from the index distribution
    torch.cuda.synchronize()
    gen_s = time.perf_counter() - t0

    # warmup on a small slice
    knn(x[: args.index_chunk], q[:1024], args.k, query_chunk=1024,
        index_chunk=args.index_chunk)
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    d, i = knn(x, q, args.k, query_chunk=args.query_chunk,
               index_chunk=args.index_chunk)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    # sanity: each query's nearest neighbor should be itself (distance 0)
    self_hit = float((i[:, 0] == torch.arange(args.queries, device=dev)).float().mean())

    print(json.dumps({
        "metric": "fused L2-NN + k-select (k=64) bf16, 100M-row index",
        "query_rows_per_sec": round(args.queries / dt, 1),
        "Gdist_per_sec": round(args.queries * args.index_rows / dt / 1e9, 1),
        "elapsed_s": round(dt, 3),
        "index_rows": args.index_rows, "dim": args.dim,
        "queries": args.queries, "k": args.k,
        "self_hit_rate": self_hit,
        "gen_s": round(gen_s, 1),
    }), flush=True)


if __name__ == "__main__":
    main()
