#!/usr/bin/env python3
"""BASELINE config 2: pairwise L2-expanded distance, 1M x 128 fp32, 1 GPU.

Metric: Gdist/s (distance values produced per second). The 1M x 1M fp32
output (4 TB) cannot be materialized in 288 GB HBM, so the benchmark computes
the full matrix in output tiles (the production access pattern for anything
consuming tiles, e.g. knn): per tile GEMM (engine selectable) + fused
L2 epilogue, writing into a reusable tile buffer.

Usage: python benchmarks/bench_pairwise.py [--rows 1000000] [--dim 128]
         [--mode native|bf16x2|bf16x3] [--tile-q 65536] [--tile-i 131072]
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
    p.add_argument("--rows", type=int, default=1_000_000)
    p.add_argument("--dim", type=int, default=128)
    p.add_argument("--mode", default="native",
                   choices=["native", "bf16x2", "bf16x3", "mfma2", "mfma3"])
    p.add_argument("--tile-q", type=int, default=50000)
    p.add_argument("--tile-i", type=int, default=100000)
    p.add_argument("--max-tiles", type=int, default=0, help="0 = full matrix")
    args = p.parse_args()

    assert torch.cuda.is_available()
    dev = torch.device("cuda")
    from raft_amd.random import make_blobs, RngState
    from raft_amd._ext import require_ext
    from raft_amd.linalg.gemm import gemm_fp32_emulated, _split_bf16

    ext = require_ext()
    assert args.rows % args.tile_q == 0 and args.rows % args.tile_i == 0, \
        "tile sizes must divide rows (keeps the output tile buffer contiguous)"
    x, _, _ = make_blobs(args.rows, args.dim, n_clusters=1000, cluster_std=1.0,
                         state=RngState(seed=1), device=dev)
    x = x.contiguous()
    xn = (x * x).sum(dim=1)

    out = torch.empty((args.tile_q, args.tile_i), dtype=torch.float32, device=dev)
    if args.mode != "native":
        nsl = 2 if args.mode in ("bf16x2", "mfma2") else 3
        x_slices = _split_bf16(x, nsl)

    def tile(q0, q1, i0, i1):
        xq = x[q0:q1]
        xi = x[i0:i1]
        g = out[: q1 - q0, : i1 - i0]
        if args.mode == "native":
            torch.matmul(xq, xi.t(), out=g)
        elif args.mode in ("mfma2", "mfma3"):
            sq = [s[q0:q1] for s in x_slices]
            si = [s[i0:i1] for s in x_slices]
            ext.pairwise_l2_mfma(sq, si, xn[q0:q1].contiguous(),
                                 xn[i0:i1].contiguous(), g)
            return
        else:
            sq = [s[q0:q1] for s in x_slices]
            si = [s[i0:i1] for s in x_slices]
            ext.gemm_bf16_f32_nt(sq[0], si[0], g, 0.0)
            ext.gemm_bf16_f32_nt(sq[0], si[1], g, 1.0)
            ext.gemm_bf16_f32_nt(sq[1], si[0], g, 1.0)
            if len(sq) == 3:
                ext.gemm_bf16_f32_nt(sq[1], si[1], g, 1.0)
                ext.gemm_bf16_f32_nt(sq[0], si[2], g, 1.0)
                ext.gemm_bf16_f32_nt(sq[2], si[0], g, 1.0)
        ext.l2_epilogue_(g, xn[q0:q1].contiguous(), xn[i0:i1].contiguous())

    # warmup
    tile(0, min(args.tile_q, args.rows), 0, min(args.tile_i, args.rows))
    torch.cuda.synchronize()

    n_dist = 0
    n_tiles = 0
    t0 = time.perf_counter()
    done = False
    for q0 in range(0, args.rows, args.tile_q):
        q1 = min(q0 + args.tile_q, args.rows)
        for i0 in range(0, args.rows, args.tile_i):
            i1 = min(i0 + args.tile_i, args.rows)
            tile(q0, q1, i0, i1)
            n_dist += (q1 - q0) * (i1 - i0)
            n_tiles += 1
            if args.max_tiles and n_tiles >= args.max_tiles:
                done = True
                break
        if done:
            break
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    print(json.dumps({
        "metric": "pairwise L2-expanded throughput (Gdist/s), 1M x 128 fp32",
        "value": round(n_dist / dt / 1e9, 3),
        "unit": "Gdist/s",
        "elapsed_s": round(dt, 3),
        "n_dist": n_dist,
        "mode": args.mode,
        "rows": args.rows, "dim": args.dim,
        "tile": [args.tile_q, args.tile_i],
        "full_matrix": not args.max_tiles,
    }), flush=True)


if __name__ == "__main__":
    main()
