"""Lanczos smallest eigenpairs of a sparse symmetric matrix (pylibraft-style).

    python examples/eigsh_example.py [--n 20000] [--k 6]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd.sparse import CSR
from raft_amd.sparse.solver import eigsh


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=20_000)
    ap.add_argument("--k", type=int, default=6)
    ap.add_argument("--nnz-per-row", type=int, default=16)
    args = ap.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    n, m = args.n, args.nnz_per_row
    rows = torch.arange(n, device=dev).repeat_interleave(m)
    cols = torch.randint(0, n, (n * m,), device=dev)
    vals = torch.randn(n * m, device=dev) * 0.1
    dense_like = torch.sparse_coo_tensor(torch.stack([rows, cols]), vals,
                                         (n, n)).coalesce()
    sym = (dense_like + dense_like.t()).coalesce() / 2
    a = CSR.from_torch_sparse(sym.to_sparse_csr())
    t0 = time.perf_counter()
    w, v = eigsh(a, k=args.k, maxiter=200, tol=1e-6)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    res = torch.stack([(a.to_torch_sparse() @ v[:, i] - w[i] * v[:, i]).norm()
                       for i in range(args.k)])
    print(f"device={dev} n={n} nnz={a.nnz}: eigsh k={args.k} in {dt:.2f}s; "
          f"eigenvalues {['%.4f' % float(x) for x in w]}; "
          f"max residual {float(res.max()):.2e}")


if __name__ == "__main__":
    main()
