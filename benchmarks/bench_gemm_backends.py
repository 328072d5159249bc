"""A/B rocBLAS gemm_ex vs hipBLASLt for the bf16->f32 building block.

Run twice: RAFT_AMD_GEMM_BACKEND=rocblas|hipblaslt (read once per process).
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import raft_amd._C as C

def main():
    torch.manual_seed(0)
    be = os.environ.get("RAFT_AMD_GEMM_BACKEND", "rocblas")
    # correctness vs fp32 torch
    a = torch.randn(512, 256, device="cuda").to(torch.bfloat16)
    b = torch.randn(256, 384, device="cuda").to(torch.bfloat16)
    ref = a.float() @ b.float()
    got = C.gemm_bf16_f32(a, b, None, 0.0)
    nt = C.gemm_bf16_f32_nt(a, b.T.contiguous(), None, 0.0)
    err = (got - ref).abs().max().item()
    errnt = (nt - ref).abs().max().item()
    print(f"[{be}] max err NN {err:.3e} NT {errnt:.3e}")
    assert err < 1e-2 and errnt < 1e-2
    # timing
    for (m, n, k) in [(8192, 8192, 8192), (10_000_000 // 128, 1024, 256)]:
        A = torch.randn(m, k, device="cuda").to(torch.bfloat16)
        B = torch.randn(k, n, device="cuda").to(torch.bfloat16)
        for _ in range(3):
            C.gemm_bf16_f32(A, B, None, 0.0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        it = 10
        for _ in range(it):
            C.gemm_bf16_f32(A, B, None, 0.0)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / it
        tf = 2 * m * n * k / dt / 1e12
        print(f"[{be}] {m}x{n}x{k}: {dt*1e3:.3f} ms = {tf:.0f} TF bf16")

main()
