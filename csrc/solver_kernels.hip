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

// ---------------------------------------------------------------------------
// Fused Lanczos step fragments (the eager loop is host-dispatch bound:
// ~16 eager ops per step at 10M rows; hipGraph capture regressed on ROCm
// 7.2 — see sparse/solver/lanczos.py). Each kernel folds several vector
// ops + a scalar reduction into ONE launch; scalars live in device memory
// (alpha/beta/norm2) so no host sync occurs inside the recurrence.
// ---------------------------------------------------------------------------

// u -= beta * v_prev (beta read from t_mat device memory; skipped if null);
// partial dot(v_i, u) accumulated into alpha_out[0] (pre-zeroed, fp64).
__global__ void lanczos_pre_kernel(float* __restrict__ u,
                                   const float* __restrict__ v_i,
                                   const float* __restrict__ v_prev,
                                   const float* __restrict__ beta,
                                   double* __restrict__ alpha_out,
                                   long long n) {
  double local = 0.0;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float ui = u[i];
    if (v_prev) ui -= (*beta) * v_prev[i];
    u[i] = ui;
    local += (double)v_i[i] * ui;
  }
  for (int off = RAFT_AMD_WAVE / 2; off > 0; off >>= 1)
    local += __shfl_down(local, off, RAFT_AMD_WAVE);
  if ((threadIdx.x % RAFT_AMD_WAVE) == 0) atomicAdd(alpha_out, local);
}

// u -= alpha * v_i (alpha from device fp64); also writes alpha into
// t_diag[0] as fp32 (t_mat[i, i]).
__global__ void lanczos_sub_alpha_kernel(float* __restrict__ u,
                                         const float* __restrict__ v_i,
                                         const double* __restrict__ alpha,
                                         float* __restrict__ t_diag,
                                         long long n) {
  const float a = (float)(*alpha);
  if (blockIdx.x == 0 && threadIdx.x == 0 && t_diag) *t_diag = a;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    u[i] -= a * v_i[i];
}

// partial ||u||^2 into norm2_out[0] (pre-zeroed, fp64)
__global__ void lanczos_norm2_kernel(const float* __restrict__ u,
                                     double* __restrict__ norm2_out,
                                     long long n) {
  double local = 0.0;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    local += (double)u[i] * u[i];
  for (int off = RAFT_AMD_WAVE / 2; off > 0; off >>= 1)
    local += __shfl_down(local, off, RAFT_AMD_WAVE);
  if ((threadIdx.x % RAFT_AMD_WAVE) == 0) atomicAdd(norm2_out, local);
}

// v_next = u / max(sqrt(norm2), 1e-300); writes beta=sqrt(norm2) into the
// two t_mat couplings (either may be null at the cycle end).
__global__ void lanczos_normalize_kernel(const float* __restrict__ u,
                                         float* __restrict__ v_next,
                                         const double* __restrict__ norm2,
                                         float* __restrict__ t_up,
                                         float* __restrict__ t_dn,
                                         float* __restrict__ beta_out,
                                         long long n) {
  const double b = sqrt(*norm2);
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    if (t_up) *t_up = (float)b;
    if (t_dn) *t_dn = (float)b;
    if (beta_out) *beta_out = (float)b;
  }
  const float inv = (float)(1.0 / (b > 1e-300 ? b : 1e-300));
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    v_next[i] = u[i] * inv;
}

void launch_lanczos_pre(float* u, const float* v_i, const float* v_prev,
                        const float* beta, double* alpha_out, long long n,
                        hipStream_t stream) {
  (void)hipMemsetAsync(alpha_out, 0, sizeof(double), stream);
  const long long grid = (n + 255) / 256;
  hipLaunchKernelGGL(lanczos_pre_kernel,
                     dim3((unsigned)(grid < 2048 ? grid : 2048)), dim3(256), 0,
                     stream, u, v_i, v_prev, beta, alpha_out, n);
}

void launch_lanczos_sub_alpha(float* u, const float* v_i, const double* alpha,
                              float* t_diag, long long n, hipStream_t stream) {
  const long long grid = (n + 255) / 256;
  hipLaunchKernelGGL(lanczos_sub_alpha_kernel,
                     dim3((unsigned)(grid < 2048 ? grid : 2048)), dim3(256), 0,
                     stream, u, v_i, alpha, t_diag, n);
}

void launch_lanczos_norm2(const float* u, double* norm2_out, long long n,
                          hipStream_t stream) {
  (void)hipMemsetAsync(norm2_out, 0, sizeof(double), stream);
  const long long grid = (n + 255) / 256;
  hipLaunchKernelGGL(lanczos_norm2_kernel,
                     dim3((unsigned)(grid < 2048 ? grid : 2048)), dim3(256), 0,
                     stream, u, norm2_out, n);
}

void launch_lanczos_normalize(const float* u, float* v_next,
                              const double* norm2, float* t_up, float* t_dn,
                              float* beta_out, long long n,
                              hipStream_t stream) {
  const long long grid = (n + 255) / 256;
  hipLaunchKernelGGL(lanczos_normalize_kernel,
                     dim3((unsigned)(grid < 2048 ? grid : 2048)), dim3(256), 0,
                     stream, u, v_next, norm2, t_up, t_dn, beta_out, n);
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
