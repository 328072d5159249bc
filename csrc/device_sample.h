// Device-side sampling helpers — wave/block weighted random selection for
// use INSIDE kernels.
//
// Reference parity: raft/random/device/sample.cuh warp_random_sample (:31) /
// block_random_sample (:69), re-derived for 64-wide wavefronts: each lane
// proposes (weight, payload); the wave (or block) selects ONE payload with
// probability proportional to weight, via the exponential-race trick
// (argmin of -log(u)/w == weighted reservoir A-Res) and wave shuffle
// reductions — no LDS needed at wave scope.
#pragma once

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

// u in (0,1]; w > 0 selects, w <= 0 never selected. Returns the winning
// lane's payload to ALL lanes.
__device__ __forceinline__ int wave_random_sample(float u, float w,
                                                  int payload) {
  float key = w > 0.f ? -__logf(u) / w : INFINITY;  // smallest key wins
  int best = payload;
#pragma unroll
  for (int off = RAFT_AMD_WAVE / 2; off > 0; off >>= 1) {
    const float ok = __shfl_xor(key, off, RAFT_AMD_WAVE);
    const int op = __shfl_xor(best, off, RAFT_AMD_WAVE);
    if (ok < key || (ok == key && op < best)) {
      key = ok;
      best = op;
    }
  }
  return best;
}

// block-scope: wave winners race through LDS (caller provides >= 2*NW floats
// + NW ints of scratch; BLOCK threads, BLOCK % 64 == 0). Returns the winning
// payload to ALL threads.
template <int BLOCK>
__device__ __forceinline__ int block_random_sample(float u, float w,
                                                   int payload, float* sh_key,
                                                   int* sh_payload) {
  constexpr int NW = BLOCK / 64;
  float key = w > 0.f ? -__logf(u) / w : INFINITY;
  int best = payload;
#pragma unroll
  for (int off = RAFT_AMD_WAVE / 2; off > 0; off >>= 1) {
    const float ok = __shfl_xor(key, off, RAFT_AMD_WAVE);
    const int op = __shfl_xor(best, off, RAFT_AMD_WAVE);
    if (ok < key || (ok == key && op < best)) {
      key = ok;
      best = op;
    }
  }
  const int wid = threadIdx.x / RAFT_AMD_WAVE;
  if ((threadIdx.x % RAFT_AMD_WAVE) == 0) {
    sh_key[wid] = key;
    sh_payload[wid] = best;
  }
  __syncthreads();
  float bk = sh_key[0];
  int bp = sh_payload[0];
#pragma unroll
  for (int wv = 1; wv < NW; wv++) {
    if (sh_key[wv] < bk || (sh_key[wv] == bk && sh_payload[wv] < bp)) {
      bk = sh_key[wv];
      bp = sh_payload[wv];
    }
  }
  __syncthreads();
  return bp;
}

}  // namespace raft_amd
