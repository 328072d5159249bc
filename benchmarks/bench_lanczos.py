#!/usr/bin/env python3
"""BASELINE config 4: CSR SpMV + Lanczos top-k eigenpairs, 10M x 10M nnz~1e8.

Two metrics:
  * raw SpMV sweeps/sec (the HBM-bound inner primitive, ~12 B/nnz)
  * Lanczos iterations/sec (SpMV + dot/axpy + 2-gemv full reorth per iter)

Synthetic symmetric graph: ~nnz_per_row random neighbors per row, symmetrized
(deterministic from seed). Generated on-device.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_symmetric_csr(n, nnz_per_row, device, seed=0):
    from raft_amd.sparse.types import COO
    from raft_amd.sparse.convert import coo_to_csr
    torch.manual_seed(seed)   # deterministic graph
    m = n * nnz_per_row // 2
    rows = torch.randint(0, n, (m,), device=device, dtype=torch.int64)
    cols = torch.randint(0, n, (m,), device=device, dtype=torch.int64)
    vals = torch.rand(m, device=device, dtype=torch.float32) + 0.1
    rows2 = torch.cat([rows, cols])
    cols2 = torch.cat([cols, rows])
    vals2 = torch.cat([vals, vals])
    # sort by (row, col); keep duplicates (they just add)
    key = rows2 * n + cols2
    order = torch.argsort(key)
    coo = COO(rows2[order], cols2[order], vals2[order], n, n)
    from raft_amd.sparse.convert import sorted_coo_to_csr
    csr = sorted_coo_to_csr(coo)
    # add diagonal dominance so the spectrum is well-behaved
    return csr


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=10_000_000)
    p.add_argument("--nnz-per-row", type=int, default=10)
    p.add_argument("--k", type=int, default=8)
    p.add_argument("--spmv-iters", type=int, default=50)
    p.add_argument("--lanczos-restarts", type=int, default=2)
    args = p.parse_args()

    assert torch.cuda.is_available()
    dev = torch.device("cuda")
    from raft_amd.sparse.linalg import spmv
    from raft_amd.sparse.solver.lanczos import lanczos_min_eigenpairs, LanczosConfig

    t0 = time.perf_counter()
    csr = make_symmetric_csr(args.n, args.nnz_per_row, dev)
    csr.indptr = csr.indptr.to(torch.int32)
    csr.indices = csr.indices.to(torch.int32)
    torch.cuda.synchronize()
    gen_s = time.perf_counter() - t0
    nnz = csr.nnz

    # ---- raw SpMV sweeps ---------------------------------------------------
    x = torch.rand(args.n, device=dev)
    y = spmv(csr, x)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.spmv_iters):
        y = spmv(csr, x)
    torch.cuda.synchronize()
    spmv_dt = (time.perf_counter() - t0) / args.spmv_iters
    spmv_gbps = (nnz * 12 + args.n * 16) / spmv_dt / 1e9

    # vendor comparison: rocSPARSE csrmv via torch.sparse
    ts = csr.to_torch_sparse()
    xu = x.unsqueeze(1)
    _ = ts @ xu
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.spmv_iters):
        _ = ts @ xu
    torch.cuda.synchronize()
    rocsparse_dt = (time.perf_counter() - t0) / args.spmv_iters

    # ---- Lanczos ------------------------------------------------------------
    ncv = max(2 * args.k + 1, 32)
    cfg = LanczosConfig(n_components=args.k, max_iterations=args.lanczos_restarts,
                        ncv=ncv, tolerance=0.0)  # fixed work: no early exit
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    w, v = lanczos_min_eigenpairs(csr, k=args.k, config=cfg)
    torch.cuda.synchronize()
    lancz_dt = time.perf_counter() - t0
    # total Lanczos steps executed: ncv + restarts * (ncv - k)
    steps = ncv + args.lanczos_restarts * (ncv - args.k)

    print(json.dumps({
        "metric": "CSR SpMV + Lanczos, 10M x 10M nnz~1e8 fp32",
        "spmv_ms": round(spmv_dt * 1e3, 3),
        "spmv_sweeps_per_sec": round(1.0 / spmv_dt, 2),
        "spmv_effective_GBps": round(spmv_gbps, 1),
        "rocsparse_spmv_ms": round(rocsparse_dt * 1e3, 3),
        "lanczos_steps_per_sec": round(steps / lancz_dt, 2),
        "lanczos_elapsed_s": round(lancz_dt, 2),
        "lanczos_steps": steps,
        "eigenvalues_head": [round(float(t), 5) for t in w[:4]],
        "n": args.n, "nnz": nnz, "k": args.k, "ncv": ncv,
        "gen_s": round(gen_s, 2),
    }), flush=True)


if __name__ == "__main__":
    main()
