// Counter-based RNG (PCG32 XSH-RR) + fused make_blobs generator.
//
// Reference parity (WHAT): raft/random/rng_device.cuh PCGenerator (:536) +
// grid-stride rngKernel (:680) with per-thread subsequences; make_blobs's
// single fused generate_data_kernel (detail/make_blobs.cuh:88).
//
// The (seed, subsequence, flat-index) -> u32 mapping here matches
// raft_amd/random/rng.py::_pcg32_block EXACTLY, so CPU and GPU draws are
// bitwise identical — tests/test_gpu_kernels.py asserts this.

#include <hip/hip_runtime.h>

#include "common.h"
#include "device_sample.h"

namespace raft_amd {

__device__ __forceinline__ uint32_t pcg32_hash(uint64_t seed, uint64_t subseq,
                                               uint64_t idx) {
  const uint64_t MULT = 6364136223846793005ull;
  const uint64_t inc = (subseq << 1) | 1ull;
  uint64_t state = (idx + seed) * MULT + inc;
  state = state * MULT + inc;
  state = state * MULT + inc;
  const uint32_t xorshifted = (uint32_t)(((state >> 18) ^ state) >> 27);
  const uint32_t rot = (uint32_t)(state >> 59);
  return (xorshifted >> rot) | (xorshifted << ((32u - rot) & 31u));
}

// Philox4x32-10 (reference rng_device.cuh PhiloxGenerator :426): counter =
// (idx_lo, idx_hi, subseq_lo, subseq_hi), key = seed; returns x0. Matches
// raft_amd/random/rng.py::_philox_block bitwise.
__device__ __forceinline__ uint32_t philox_hash(uint64_t seed, uint64_t subseq,
                                                uint64_t idx) {
  uint32_t c0 = (uint32_t)idx, c1 = (uint32_t)(idx >> 32);
  uint32_t c2 = (uint32_t)subseq, c3 = (uint32_t)(subseq >> 32);
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; r++) {
    const uint64_t p0 = (uint64_t)c0 * 0xD2511F53u;
    const uint64_t p1 = (uint64_t)c2 * 0xCD9E8D57u;
    const uint32_t n0 = (uint32_t)(p1 >> 32) ^ c1 ^ k0;
    const uint32_t n1 = (uint32_t)p1;
    const uint32_t n2 = (uint32_t)(p0 >> 32) ^ c3 ^ k1;
    const uint32_t n3 = (uint32_t)p0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  return c0;
}

__device__ __forceinline__ uint32_t rng_hash(uint64_t seed, uint64_t subseq,
                                             uint64_t idx, int gen) {
  return gen == 1 ? philox_hash(seed, subseq, idx)
                  : pcg32_hash(seed, subseq, idx);
}

__device__ __forceinline__ float u32_to_f01(uint32_t u) {
  // matches the python oracle: (u + 0.5) / 2^32 computed in double
  return (float)(((double)u + 0.5) * (1.0 / 4294967296.0));
}

__global__ void rng_uniform_kernel(float* __restrict__ out, long long n,
                                   uint64_t seed, uint64_t subseq, int gen) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    out[i] = u32_to_f01(rng_hash(seed, subseq, (uint64_t)i, gen));
}

__device__ __forceinline__ float box_muller(uint64_t seed, uint64_t subseq,
                                            uint64_t idx, int gen = 0) {
  const double u1 = ((double)rng_hash(seed, subseq, idx, gen) + 0.5) * (1.0 / 4294967296.0);
  const double u2 = ((double)rng_hash(seed, subseq + 1, idx, gen) + 0.5) * (1.0 / 4294967296.0);
  const double r = sqrt(-2.0 * log(u1));
  return (float)(r * cos(2.0 * M_PI * u2));
}

__global__ void rng_normal_kernel(float* __restrict__ out, long long n,
                                  uint64_t seed, uint64_t subseq, int gen) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    out[i] = box_muller(seed, subseq, (uint64_t)i, gen);
}

// fused make_blobs: labels (subseq) + gaussian offsets (subseq+1, +2) + center
// add, one write pass. centers [k, d] stream from L2 (tiny).
__global__ void make_blobs_kernel(float* __restrict__ x, int* __restrict__ labels,
                                  const float* __restrict__ centers, long long n_rows,
                                  long long d, int k, float std, uint64_t seed,
                                  uint64_t subseq) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  const long long total = n_rows * d;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < total; t += stride) {
    const long long i = t / d;
    const long long j = t % d;
    const int lab = (int)(pcg32_hash(seed, subseq, (uint64_t)i) % (uint32_t)k);
    if (j == 0) labels[i] = lab;
    const float z = box_muller(seed, subseq + 1, (uint64_t)t);
    x[t] = centers[(long long)lab * d + j] + std * z;
  }
}

void launch_rng_uniform(float* out, long long n, uint64_t seed, uint64_t subseq,
                        hipStream_t stream, int gen) {
  hipLaunchKernelGGL(rng_uniform_kernel, dim3(grid_1d(n, 256)), dim3(256), 0, stream,
                     out, n, seed, subseq, gen);
}

void launch_rng_normal(float* out, long long n, uint64_t seed, uint64_t subseq,
                       hipStream_t stream, int gen) {
  hipLaunchKernelGGL(rng_normal_kernel, dim3(grid_1d(n, 256)), dim3(256), 0, stream,
                     out, n, seed, subseq, gen);
}

void launch_make_blobs(float* x, int* labels, const float* centers, long long n_rows,
                       long long d, int k, float std, uint64_t seed, uint64_t subseq,
                       hipStream_t stream) {
  hipLaunchKernelGGL(make_blobs_kernel, dim3(grid_1d(n_rows * d, 256)), dim3(256), 0,
                     stream, x, labels, centers, n_rows, d, k, std, seed, subseq);
}

}  // namespace raft_amd

namespace raft_amd {

// test driver for the device sampling helpers: each block draws ONE index in
// [0, 256) proportional to weights[] via block_random_sample; out[b] = draw.
__global__ void device_sample_test_kernel(const float* __restrict__ weights,
                                          int* __restrict__ out, int n_draws,
                                          uint64_t seed) {
  __shared__ float sh_key[4];
  __shared__ int sh_payload[4];
  const int b = blockIdx.x;
  if (b >= n_draws) return;
  const int tid = threadIdx.x;  // 256 threads = 256 candidate payloads
  const float u =
      ((double)rng_hash(seed, (uint64_t)b, (uint64_t)tid, 0) + 0.5) *
      (1.0 / 4294967296.0);
  const int pick = block_random_sample<256>(u, weights[tid], tid, sh_key,
                                            sh_payload);
  if (tid == 0) out[b] = pick;
}

void launch_device_sample_test(const float* weights, int* out, int n_draws,
                               uint64_t seed, hipStream_t stream) {
  hipLaunchKernelGGL(device_sample_test_kernel, dim3(n_draws), dim3(256), 0,
                     stream, weights, out, n_draws, seed);
}

}  // namespace raft_amd
