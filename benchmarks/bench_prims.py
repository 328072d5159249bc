"""Primitive micro-benchmark suite (reference parity: cpp/bench/prims/* —
google-benchmark harness over linalg/matrix/random/sparse prims).

Prints one line per (prim, shape): mean ms over `--iters` timed runs after
warmup, plus an effective-bandwidth or throughput figure where meaningful.

Run on a GPU box:  python benchmarks/bench_prims.py [--iters 20]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def timeit(fn, iters, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def gb(nbytes, sec):
    return nbytes / sec / 1e9


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    it = args.iters
    dev = "cuda"
    torch.manual_seed(0)
    rows = []

    def rec(name, shape, ms, extra=""):
        rows.append(f"{name:<28} {shape:<24} {ms*1e3:9.3f} ms  {extra}")
        print(rows[-1], flush=True)

    from raft_amd import linalg, matrix, stats, random as ramd, sparse
    from raft_amd.linalg import Apply
    from raft_amd.linalg import reduce_rows_by_key

    # --- linalg reductions (bench/prims/linalg/reduce.cu) ---
    for (m, n) in [(1 << 20, 64), (1 << 16, 1024), (1 << 22, 16)]:
        x = torch.randn(m, n, device=dev)
        t = timeit(lambda: linalg.reduce(x, Apply.ALONG_ROWS), it)
        rec("reduce rows(add)", f"{m}x{n}", t, f"{gb(x.numel()*4, t):.0f} GB/s")
        t = timeit(lambda: linalg.reduce(x, Apply.ALONG_COLUMNS), it)
        rec("reduce cols(add)", f"{m}x{n}", t, f"{gb(x.numel()*4, t):.0f} GB/s")

    x = torch.randn(1 << 20, 256, device=dev)
    t = timeit(lambda: linalg.normalize(x), it)
    rec("normalize(L2)", "1Mx256", t, f"{gb(x.numel()*8, t):.0f} GB/s")

    v = torch.randn(256, device=dev)
    t = timeit(lambda: linalg.matrix_vector_op(x, v, op="add"), it)
    rec("matrix_vector_op(add)", "1Mx256", t, f"{gb(x.numel()*8, t):.0f} GB/s")

    # --- reduce_rows_by_key (bench/prims/linalg/reduce_rows_by_key.cu) ---
    for m in (1 << 20, 10_000_000):
        xk = torch.randn(m, 256, device=dev)
        keys = torch.randint(0, 1024, (m,), device=dev, dtype=torch.int32)
        t = timeit(lambda: reduce_rows_by_key(xk, keys, n_keys=1024), it)
        rec("reduce_rows_by_key", f"{m}x256 k=1024", t,
            f"{gb(xk.numel()*4, t):.0f} GB/s")
        del xk

    # --- select_k grid (bench/prims/matrix/select_k.cu shapes) ---
    for (b, n, k) in [(20000, 500, 32), (1000, 10000, 64), (100, 100000, 64),
                      (10, 1000000, 128), (8192, 1000000, 64)]:
        if b * n > (1 << 31):
            continue
        xs = torch.randn(b, n, device=dev)
        t = timeit(lambda: matrix.select_k(xs, k), max(3, it // 4))
        rec("select_k", f"[{b}x{n}] k={k}", t, f"{gb(xs.numel()*4, t):.0f} GB/s")
        del xs

    # --- random (bench/prims/random/*) ---
    from raft_amd.random import RngState
    t = timeit(lambda: ramd.uniform((1 << 24,), state=RngState(1), device=dev), it)
    rec("rng uniform", "16M", t, f"{(1<<24)/t/1e9:.1f} Gsamp/s")
    t = timeit(lambda: ramd.normal((1 << 24,), state=RngState(1), device=dev), it)
    rec("rng normal", "16M", t, f"{(1<<24)/t/1e9:.1f} Gsamp/s")
    t = timeit(lambda: ramd.make_blobs(1 << 20, 64, n_clusters=32,
                                       state=RngState(2), device=dev), max(3, it // 2))
    rec("make_blobs", "1Mx64 c=32", t)
    t = timeit(lambda: ramd.permute_rows(x[:, :64]), it)
    rec("permute rows", "1Mx64", t, f"{gb((1<<20)*64*8, t):.0f} GB/s")

    # --- more reductions / norms ---
    import raft_amd._C as C
    t = timeit(lambda: C.row_argmin(x), it)
    rec("row_argmin", "1Mx256", t, f"{gb(x.numel()*4, t):.0f} GB/s")
    xb = x.bfloat16().contiguous()
    t = timeit(lambda: C.rows_sqnorm_bf16(xb), it)
    rec("rows_sqnorm_bf16", "1Mx256", t, f"{gb(x.numel()*2, t):.0f} GB/s")
    vr = torch.randn(1 << 20, device=dev)
    t = timeit(lambda: linalg.matrix_vector_op(x, vr, op="add", along_rows=False), it)
    rec("matrix_vector_op(cols)", "1Mx256", t, f"{gb(x.numel()*8, t):.0f} GB/s")

    # --- matrix ops ---
    t = timeit(lambda: matrix.gather(x, torch.randint(0, 1 << 20, (1 << 19,),
                                                      device=dev)), it)
    rec("gather rows", "512Kof1M x256", t, f"{gb((1<<19)*256*8, t):.0f} GB/s")

    # --- stats (bench/prims/stats-ish) ---
    t = timeit(lambda: stats.meanvar(x), it)
    rec("meanvar", "1Mx256", t, f"{gb(x.numel()*4, t):.0f} GB/s")
    t = timeit(lambda: stats.histogram(x[:, 0].contiguous(), n_bins=256), it)
    rec("histogram", "1M bins=256", t)
    xs_small = torch.randn(100000, 64, device=dev)
    t = timeit(lambda: stats.cov(xs_small), it)
    rec("cov", "100Kx64", t)

    # --- sparse (bench/prims/sparse/*) ---
    m = 1 << 20
    nnz_per = 32
    cols = torch.randint(0, m, (m * nnz_per,), device=dev, dtype=torch.int32)
    indptr = torch.arange(0, m * nnz_per + 1, nnz_per, device=dev,
                          dtype=torch.int32)
    vals = torch.randn(m * nnz_per, device=dev)
    a = sparse.CSR(indptr, cols, vals, m, m)
    xv = torch.randn(m, device=dev)
    t = timeit(lambda: sparse.spmv(a, xv), it)
    rec("csr spmv", f"{m} rows {m*nnz_per} nnz", t,
        f"{gb(m*nnz_per*8 + m*8, t):.0f} GB/s")

    print("\n".join(["", "=== summary (commit to profiles/) ==="] + rows))


main()
