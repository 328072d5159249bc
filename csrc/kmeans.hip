// k-means centroid update: reduce_rows_by_key via wave-per-row atomics.
//
// Reference parity (WHAT): raft/linalg/reduce_rows_by_key (detail, smem-binned
// + atomic kernels). MI355X design: one wave per input row; lanes stride the
// feature dim with float4 vector loads and issue device-scope fp32 atomicAdds
// into sums[key]. With k in the hundreds+ the per-key contention is low and
// this is HBM-bound at ~2 passes over X (guide G12: pre-aggregate per block is
// unnecessary at this contention level; measured before optimizing further).

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

__global__ void reduce_rows_by_key_kernel(const float* __restrict__ x,
                                          const int* __restrict__ keys,
                                          float* __restrict__ sums,
                                          long long n_rows, long long d) {
  const long long waves_per_block = blockDim.x / RAFT_AMD_WAVE;
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  long long row = (long long)blockIdx.x * waves_per_block + threadIdx.x / RAFT_AMD_WAVE;
  const long long stride = (long long)gridDim.x * waves_per_block;
  for (; row < n_rows; row += stride) {
    const int key = keys[row];
    const float* rp = x + row * d;
    float* sp = sums + (long long)key * d;
    const long long d4 = d / 4;
    const float4* rp4 = reinterpret_cast<const float4*>(rp);
    for (long long j = lane; j < d4; j += RAFT_AMD_WAVE) {
      float4 v = rp4[j];
      atomicAdd(&sp[j * 4 + 0], v.x);
      atomicAdd(&sp[j * 4 + 1], v.y);
      atomicAdd(&sp[j * 4 + 2], v.z);
      atomicAdd(&sp[j * 4 + 3], v.w);
    }
    for (long long j = d4 * 4 + lane; j < d; j += RAFT_AMD_WAVE) atomicAdd(&sp[j], rp[j]);
  }
}

void launch_reduce_rows_by_key(const float* x, const int* keys, float* sums,
                               long long n_rows, long long d, hipStream_t stream) {
  int grid = grid_1d(n_rows * RAFT_AMD_WAVE, 256);
  hipLaunchKernelGGL(reduce_rows_by_key_kernel, dim3(grid), dim3(256), 0, stream,
                     x, keys, sums, n_rows, d);
}

}  // namespace raft_amd
