// Small solver kernels.
//
// cholesky_r1_update: rank-1 Cholesky update (hyperbolic rotations) as ONE
// kernel launch. Reference parity: raft/linalg/cholesky_r1_update.cuh. The
// k-recurrence is inherently serial, but each step's column update is
// parallel — the round-1 Python loop issued ~6 kernel launches per k
// (n launch round-trips); here a single block walks k
// with the column update strided across threads.

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

template <typename T>
__global__ void cholesky_r1_update_kernel(T* __restrict__ l, T* __restrict__ x,
                                          int n, long long ld) {
  __shared__ T sh_c, sh_s;
  const int tid = threadIdx.x;
  for (int k = 0; k < n; k++) {
    if (tid == 0) {
      const T lkk = l[(long long)k * ld + k];
      const T xk = x[k];
      const T r = sqrt(lkk * lkk + xk * xk);
      sh_c = r / lkk;
      sh_s = xk / lkk;
      l[(long long)k * ld + k] = r;
    }
    __syncthreads();
    const T c = sh_c, s = sh_s;
    for (int i = k + 1 + tid; i < n; i += blockDim.x) {
      const T li = (l[(long long)i * ld + k] + s * x[i]) / c;
      l[(long long)i * ld + k] = li;
      x[i] = c * x[i] - s * li;
    }
    __syncthreads();
  }
}

void launch_cholesky_r1_update_f32(float* l, float* x, int n, long long ld,
                                   hipStream_t stream) {
  hipLaunchKernelGGL((cholesky_r1_update_kernel<float>), dim3(1), dim3(256), 0,
                     stream, l, x, n, ld);
}

void launch_cholesky_r1_update_f64(double* l, double* x, int n, long long ld,
                                   hipStream_t stream) {
  hipLaunchKernelGGL((cholesky_r1_update_kernel<double>), dim3(1), dim3(256),
                     0, stream, l, x, n, ld);
}

}  // namespace raft_amd
