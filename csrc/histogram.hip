// Per-column histograms with strategy dispatch by n_bins.
//
// Reference parity: raft/stats/detail/histogram.cuh — gmem atomics (:69),
// smem (:102), smem-bits packed counters (:196), hash (:240) auto-chosen by
// nbins (histogram.cuh:52-85).
//
// MI355X design (wave64/LDS-first):
//   * LDS-MULTI  (nbins <= 2048): one LDS sub-histogram PER WAVE
//     (NW x nbins u32 <= 64 KiB) — wave-private counters kill the LDS atomic
//     contention that a single shared histogram has at small nbins (the
//     reference's smem-bits strategy solves the same contention with packed
//     counters; wave-privatization is the wave64-native answer).
//   * LDS-SINGLE (nbins <= 16384): one LDS histogram per block.
//   * GMEM       (larger): device-scope atomics into the output.
// Grid: (column, row-chunk) — each block bins a row chunk of ONE column and
// merges into out[bin, col] with one global atomic per touched bin.
// Binning: linear [lo, hi) -> floor((v - lo) * scale), clamped.

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

template <int BLOCK, bool MULTI>
__global__ void histogram_lds_kernel(const float* __restrict__ x, long long n,
                                     long long d, int n_bins, float lo,
                                     float scale,
                                     unsigned long long* __restrict__ out) {
  extern __shared__ unsigned int sh[];
  const int col = blockIdx.x;
  const int chunk = blockIdx.y;
  const int nchunks = gridDim.y;
  const int wid = threadIdx.x / RAFT_AMD_WAVE;
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  const int copies = MULTI ? NW : 1;
  for (int b = threadIdx.x; b < copies * n_bins; b += BLOCK) sh[b] = 0;
  __syncthreads();
  unsigned int* mine = sh + (MULTI ? wid * n_bins : 0);
  const long long rows_per = (n + nchunks - 1) / nchunks;
  const long long r0 = (long long)chunk * rows_per;
  const long long r1 = r0 + rows_per < n ? r0 + rows_per : n;
  for (long long r = r0 + threadIdx.x; r < r1; r += BLOCK) {
    const float v = x[r * d + col];
    int b = (int)floorf((v - lo) * scale);
    b = b < 0 ? 0 : (b >= n_bins ? n_bins - 1 : b);
    atomicAdd(&mine[b], 1u);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < n_bins; b += BLOCK) {
    unsigned long long s = 0;
    for (int c = 0; c < copies; c++) s += sh[c * n_bins + b];
    if (s) atomicAdd(&out[(long long)b * d + col], s);
  }
}

template <int BLOCK>
__global__ void histogram_gmem_kernel(const float* __restrict__ x, long long n,
                                      long long d, int n_bins, float lo,
                                      float scale,
                                      unsigned long long* __restrict__ out) {
  const int col = blockIdx.x;
  const int chunk = blockIdx.y;
  const int nchunks = gridDim.y;
  const long long rows_per = (n + nchunks - 1) / nchunks;
  const long long r0 = (long long)chunk * rows_per;
  const long long r1 = r0 + rows_per < n ? r0 + rows_per : n;
  for (long long r = r0 + threadIdx.x; r < r1; r += BLOCK) {
    const float v = x[r * d + col];
    int b = (int)floorf((v - lo) * scale);
    b = b < 0 ? 0 : (b >= n_bins ? n_bins - 1 : b);
    atomicAdd(&out[(long long)b * d + col], 1ull);
  }
}

void launch_histogram(const float* x, long long n, long long d, int n_bins,
                      float lo, float hi, unsigned long long* out,
                      hipStream_t stream) {
  const float scale = (float)n_bins / (hi - lo != 0.f ? hi - lo : 1.f);
  // enough row-chunks to fill 256 CUs x 2 even for few columns
  int nchunks = (int)((512 + d - 1) / d);
  const long long rows_min = 4096;
  if ((n + nchunks - 1) / nchunks < rows_min)
    nchunks = (int)((n + rows_min - 1) / rows_min);
  if (nchunks < 1) nchunks = 1;
  dim3 grid((unsigned)d, (unsigned)nchunks);
  constexpr int BLOCK = 256;
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  if ((size_t)NW * n_bins * 4 <= 64 * 1024) {
    hipLaunchKernelGGL((histogram_lds_kernel<BLOCK, true>), grid, dim3(BLOCK),
                       (size_t)NW * n_bins * 4, stream, x, n, d, n_bins, lo,
                       scale, out);
  } else if ((size_t)n_bins * 4 <= 64 * 1024) {
    hipLaunchKernelGGL((histogram_lds_kernel<BLOCK, false>), grid, dim3(BLOCK),
                       (size_t)n_bins * 4, stream, x, n, d, n_bins, lo, scale,
                       out);
  } else {
    hipLaunchKernelGGL((histogram_gmem_kernel<BLOCK>), grid, dim3(BLOCK), 0,
                       stream, x, n, d, n_bins, lo, scale, out);
  }
}

// ---------------------------------------------------------------------------
// Bitset kernels (reference core/bitset.hpp:33 set/test/count): O(k) scatter
// of set/clear bits and a popc count — the round-1 Python Bitset.set()
// materialized a dense bool mask per call.
// ---------------------------------------------------------------------------

__global__ void bitset_set_kernel(unsigned int* __restrict__ words,
                                  const long long* __restrict__ idx,
                                  long long k, int value) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= k) return;
  const long long j = idx[i];
  const unsigned int bit = 1u << (j & 31);
  if (value)
    atomicOr(&words[j >> 5], bit);
  else
    atomicAnd(&words[j >> 5], ~bit);
}

__global__ void bitset_test_kernel(const unsigned int* __restrict__ words,
                                   const long long* __restrict__ idx,
                                   bool* __restrict__ out, long long k) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= k) return;
  const long long j = idx[i];
  out[i] = (words[j >> 5] >> (j & 31)) & 1u;
}

__global__ void bitset_count_kernel(const unsigned int* __restrict__ words,
                                    long long n_words,
                                    unsigned long long* __restrict__ out) {
  unsigned long long local = 0;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_words; i += (long long)gridDim.x * blockDim.x)
    local += __popc(words[i]);
  // wave reduce then one atomic per wave
  for (int off = RAFT_AMD_WAVE / 2; off > 0; off >>= 1)
    local += __shfl_down(local, off, RAFT_AMD_WAVE);
  if ((threadIdx.x % RAFT_AMD_WAVE) == 0) atomicAdd(out, local);
}

void launch_bitset_set(unsigned int* words, const long long* idx, long long k,
                       int value, hipStream_t stream) {
  const long long grid = (k + 255) / 256;
  hipLaunchKernelGGL(bitset_set_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, words, idx, k, value);
}

void launch_bitset_test(const unsigned int* words, const long long* idx,
                        bool* out, long long k, hipStream_t stream) {
  const long long grid = (k + 255) / 256;
  hipLaunchKernelGGL(bitset_test_kernel, dim3((unsigned)grid), dim3(256), 0,
                     stream, words, idx, out, k);
}

void launch_bitset_count(const unsigned int* words, long long n_words,
                         unsigned long long* out, hipStream_t stream) {
  const long long grid = (n_words + 255) / 256;
  hipLaunchKernelGGL(bitset_count_kernel,
                     dim3((unsigned)(grid < 2048 ? grid : 2048)), dim3(256), 0,
                     stream, words, n_words, out);
}

}  // namespace raft_amd
