// Pairwise-distance epilogues + unexpanded (diff-accumulation) distances.
//
// Reference parity (WHAT): RAFT's historical pairwise_distance epilogues over
// the contraction engine (linalg/contractions.cuh) — L2-expanded fuses
// ||x||^2 + ||y||^2 - 2xy + clamp into one pass over the GEMM output;
// L1/Linf/Lp/Canberra/Hamming are diff-accumulation contractions.
//
// MI355X design: the epilogue kernels are memory-bound float4 streams
// (guide G13/G11); the unexpanded kernel is an LDS-tiled contraction sized
// for 64-lane wavefronts (32x32 output tile per 256-thread block, both
// operand panels staged through LDS in d-chunks of 32).

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

// --------------------------------------------------------------------------
// in-place fused L2 epilogue: g[i,j] = max(xn[i] + yn[j] - 2 g[i,j], 0)
// --------------------------------------------------------------------------
__global__ void l2_epilogue_kernel(float* __restrict__ g, const float* __restrict__ xn,
                                   const float* __restrict__ yn, long long m, long long n) {
  const long long total4 = m * (n / 4);
  const long long n4 = n / 4;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < total4; t += stride) {
    const long long i = t / n4;
    const long long j4 = (t % n4) * 4;
    float4* gp = reinterpret_cast<float4*>(g + i * n + j4);
    float4 v = *gp;
    const float x = xn[i];
    v.x = fmaxf(x + yn[j4 + 0] - 2.f * v.x, 0.f);
    v.y = fmaxf(x + yn[j4 + 1] - 2.f * v.y, 0.f);
    v.z = fmaxf(x + yn[j4 + 2] - 2.f * v.z, 0.f);
    v.w = fmaxf(x + yn[j4 + 3] - 2.f * v.w, 0.f);
    *gp = v;
  }
  // ragged tail columns
  const long long tail0 = n4 * 4;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       t < m * (n - tail0); t += stride) {
    const long long i = t / (n - tail0);
    const long long j = tail0 + t % (n - tail0);
    g[i * n + j] = fmaxf(xn[i] + yn[j] - 2.f * g[i * n + j], 0.f);
  }
}

void launch_l2_epilogue(float* g, const float* xn, const float* yn,
                        long long m, long long n, hipStream_t stream) {
  int grid = grid_1d(m * ((n + 3) / 4), 256);
  hipLaunchKernelGGL(l2_epilogue_kernel, dim3(grid), dim3(256), 0, stream, g, xn, yn, m, n);
}

// --------------------------------------------------------------------------
// fused L2-NN epilogue over a GEMM tile: per row of g [m, n], find
// argmin_j (xn[i] + yn[j] - 2 g[i,j]) without materializing distances.
// One wave per row (n is the centroid count — typically <= a few thousand).
// --------------------------------------------------------------------------
__global__ void l2nn_epilogue_kernel(const float* __restrict__ g,
                                     const float* __restrict__ xn,
                                     const float* __restrict__ yn,
                                     float* __restrict__ dmin, int* __restrict__ amin,
                                     long long m, long long n) {
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const long long waves_per_block = blockDim.x / RAFT_AMD_WAVE;
  long long row = (long long)blockIdx.x * waves_per_block + threadIdx.x / RAFT_AMD_WAVE;
  const long long stride = (long long)gridDim.x * waves_per_block;
  for (; row < m; row += stride) {
    const float* gp = g + row * n;
    float best = INFINITY;
    int bi = 0;
    for (long long j = lane; j < n; j += RAFT_AMD_WAVE) {
      float v = yn[j] - 2.f * gp[j];
      if (v < best) { best = v; bi = (int)j; }
    }
    wave_reduce_argmin(best, bi);
    if (lane == 0) {
      dmin[row] = fmaxf(best + xn[row], 0.f);
      amin[row] = bi;
    }
  }
}

void launch_l2nn_epilogue(const float* g, const float* xn, const float* yn,
                          float* dmin, int* amin, long long m, long long n,
                          hipStream_t stream) {
  int grid = grid_1d(m * RAFT_AMD_WAVE, 256);
  hipLaunchKernelGGL(l2nn_epilogue_kernel, dim3(grid), dim3(256), 0, stream,
                     g, xn, yn, dmin, amin, m, n);
}

// --------------------------------------------------------------------------
// unexpanded pairwise distances: LDS-tiled contraction.
// codes: 0=L1, 1=Linf, 2=Lp, 3=Canberra, 4=Hamming
// --------------------------------------------------------------------------
template <int CODE>
__device__ __forceinline__ float acc_op(float a, float xv, float yv, float p) {
  const float diff = xv - yv;
  if constexpr (CODE == 0) return a + fabsf(diff);
  if constexpr (CODE == 1) return fmaxf(a, fabsf(diff));
  if constexpr (CODE == 2) return a + powf(fabsf(diff), p);
  if constexpr (CODE == 3) {
    const float den = fabsf(xv) + fabsf(yv);
    return a + (den > 0.f ? fabsf(diff) / den : 0.f);
  }
  if constexpr (CODE == 4) return a + (diff != 0.f ? 1.f : 0.f);
  return a;
}

template <int CODE, int TM = 32, int TN = 32, int TK = 32>
__global__ void pairwise_unexp_kernel(const float* __restrict__ x, const float* __restrict__ y,
                                      float* __restrict__ out, long long m, long long n,
                                      long long d, float p) {
  // 256 threads -> 32x32 output tile, 4 outputs per thread (8 rows of 32)
  __shared__ float xs[TM][TK + 1];
  __shared__ float ys[TN][TK + 1];
  const long long bi = (long long)blockIdx.y * TM;  // x-row base
  const long long bj = (long long)blockIdx.x * TN;  // y-row base
  const int tx = threadIdx.x % TN;                  // output col
  const int ty0 = threadIdx.x / TN;                 // 0..7
  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  if constexpr (CODE == 1) { acc[0] = acc[1] = acc[2] = acc[3] = 0.f; }

  for (long long k0 = 0; k0 < d; k0 += TK) {
    // stage x tile [TM][TK] and y tile [TN][TK]; 256 threads load 32x32 each
    const int lr = threadIdx.x / TK;   // 0..7 rows at a time
    const int lc = threadIdx.x % TK;
    for (int r = lr; r < TM; r += 8) {
      const long long gi = bi + r;
      xs[r][lc] = (gi < m && k0 + lc < d) ? x[gi * d + k0 + lc] : 0.f;
    }
    for (int r = lr; r < TN; r += 8) {
      const long long gj = bj + r;
      ys[r][lc] = (gj < n && k0 + lc < d) ? y[gj * d + k0 + lc] : 0.f;
    }
    __syncthreads();
    const int kmax = (int)((d - k0) < TK ? (d - k0) : TK);
    for (int rr = 0; rr < 4; rr++) {
      const int row = ty0 + rr * 8;
      for (int k = 0; k < kmax; k++) {
        acc[rr] = acc_op<CODE>(acc[rr], xs[row][k], ys[tx][k], p);
      }
    }
    __syncthreads();
  }
  for (int rr = 0; rr < 4; rr++) {
    const long long gi = bi + ty0 + rr * 8;
    const long long gj = bj + tx;
    if (gi < m && gj < n) {
      float v = acc[rr];
      if constexpr (CODE == 2) v = powf(v, 1.f / p);
      if constexpr (CODE == 4) v = v / (float)d;
      out[gi * n + gj] = v;
    }
  }
}

void launch_pairwise_unexpanded(const float* x, const float* y, float* out,
                                long long m, long long n, long long d, int code,
                                float p, hipStream_t stream) {
  dim3 grid((unsigned)((n + 31) / 32), (unsigned)((m + 31) / 32));
  dim3 block(256);
  switch (code) {
    case 0: hipLaunchKernelGGL((pairwise_unexp_kernel<0>), grid, block, 0, stream, x, y, out, m, n, d, p); break;
    case 1: hipLaunchKernelGGL((pairwise_unexp_kernel<1>), grid, block, 0, stream, x, y, out, m, n, d, p); break;
    case 2: hipLaunchKernelGGL((pairwise_unexp_kernel<2>), grid, block, 0, stream, x, y, out, m, n, d, p); break;
    case 3: hipLaunchKernelGGL((pairwise_unexp_kernel<3>), grid, block, 0, stream, x, y, out, m, n, d, p); break;
    case 4: hipLaunchKernelGGL((pairwise_unexp_kernel<4>), grid, block, 0, stream, x, y, out, m, n, d, p); break;
    default: throw std::runtime_error("bad pairwise code");
  }
}

}  // namespace raft_amd
