"""Host-side attribution of the Lanczos step wall time (GPU box).

End-to-end eigsh on the config-4-shaped matrix (10M rows, 1e8 nnz), then
per-phase synchronized timings of the step's building blocks to locate the
wall-minus-kernel gap (2.6 ms GPU work vs 6.2 ms wall).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd._ext import require_ext
from raft_amd.sparse import CSR
from raft_amd.sparse.solver import eigsh


def build_csr(n=10_000_000, m=10):
    dev = "cuda"
    g = torch.Generator(device=dev).manual_seed(0)
    # structured CSR directly (m nnz/row, sorted cols per row): no coalesce
    indptr = torch.arange(0, (n + 1) * m, m, device=dev, dtype=torch.int32)
    cols = torch.randint(0, n, (n * m,), generator=g, device=dev,
                         dtype=torch.int32).view(n, m).sort(dim=1).values
    vals = torch.rand(n * m, generator=g, device=dev)
    return CSR(indptr, cols.reshape(-1).contiguous(), vals, n, n)


def main():
    ext = require_ext()
    a = build_csr()
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    w, v = eigsh(a, k=6, ncv=32, maxiter=20, tol=0.0)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    n_steps = 20 * (32 - 6) + 32
    print(f"eigsh wall {dt:.3f}s ~{n_steps} steps -> {n_steps / dt:.1f} steps/s")

    n = a.n_rows
    u = torch.rand(n, device="cuda")
    vi = torch.rand(n, device="cuda")
    basis = torch.rand(8, n, device="cuda")
    t_small = torch.zeros(32, 32, device="cuda")
    scal = torch.zeros(1, dtype=torch.float64, device="cuda")
    indptr = a.indptr.to(torch.int32).contiguous()
    indices = a.indices.to(torch.int32).contiguous()

    def bench(name, fn, iters=50):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        print(f"{name:28s} {(time.perf_counter() - t) / iters * 1e3:7.3f} ms")

    bench("spmv", lambda: ext.csr_spmv(indptr, indices, a.values, u, n))
    bench("CGS pass (matvec+addmv) x2", lambda: [
        torch.addmv(u, basis.t(), basis @ u, alpha=-1.0) for _ in range(2)])
    bench("fused pre+alpha", lambda: (
        ext.lanczos_pre_(u, vi, None, None, scal),
        ext.lanczos_sub_alpha_(u, vi, scal, t_small[0, 0:1])))
    bench("norm2+normalize", lambda: (
        ext.lanczos_norm2_(u, scal),
        ext.lanczos_normalize_(u, vi, scal, t_small[0, 1:2], t_small[1, 0:1],
                               None)))
    g32 = torch.rand(32, 32, device="cuda")
    t_sym = g32 + g32.t() + 32 * torch.eye(32, device="cuda")
    bench("eigh 32x32 GPU (1x/cycle)", lambda: torch.linalg.eigh(t_sym))

    def eigh_cpu():
        w_, s_ = torch.linalg.eigh(t_sym.cpu())
        return w_.cuda(), s_.cuda()
    bench("eigh 32x32 via CPU roundtrip", eigh_cpu)
    vfull = torch.rand(32, n, device="cuda")
    bench("ritz restart gemm (1x/cycle)", lambda: t_sym[:, :6].t() @ vfull,
          iters=20)


if __name__ == "__main__":
    main()
