// Common device helpers for raft_amd CDNA4 (gfx950) kernels.
//
// Design references: /opt/skills/guides/cdna_hip_programming.md — wave64
// idioms, vectorized IO (G13), LDS layout (G3/G4), grid sizing (G11).
// Reference-parity notes cite rapidsai/raft headers (util/reduction.cuh etc.)
// for WHAT is computed; the wave-level HOW here is CDNA4-native (64-lane
// shuffles, 2-cycle SIMD-32 issue, no 32-lane assumptions anywhere).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define RAFT_AMD_WAVE 64

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " __FILE__ ":" +   \
                               std::to_string(__LINE__));                      \
    }                                                                          \
  } while (0)

namespace raft_amd {

// ---------------------------------------------------------------------------
// wave-level reductions (reference parity: raft/util/reduction.cuh warpReduce
// — rebuilt for 64-lane wavefronts; xor-shuffle butterfly, log2(64)=6 steps)
// ---------------------------------------------------------------------------

template <typename T>
__device__ __forceinline__ T wave_reduce_sum(T v, int width = RAFT_AMD_WAVE) {
  for (int off = width >> 1; off > 0; off >>= 1) v += __shfl_xor(v, off, RAFT_AMD_WAVE);
  return v;
}

template <typename T>
__device__ __forceinline__ T wave_reduce_max(T v, int width = RAFT_AMD_WAVE) {
  for (int off = width >> 1; off > 0; off >>= 1) {
    T o = __shfl_xor(v, off, RAFT_AMD_WAVE);
    v = o > v ? o : v;
  }
  return v;
}

template <typename T>
__device__ __forceinline__ T wave_reduce_min(T v, int width = RAFT_AMD_WAVE) {
  for (int off = width >> 1; off > 0; off >>= 1) {
    T o = __shfl_xor(v, off, RAFT_AMD_WAVE);
    v = o < v ? o : v;
  }
  return v;
}

// argmin over (val, idx) pairs — the kvp reduction used by argmin/L2-NN
// (reference: raft/core/kvp.hpp + argmin_op). Tie-break: lowest index.
template <typename T, typename I>
__device__ __forceinline__ void wave_reduce_argmin(T& v, I& i, int width = RAFT_AMD_WAVE) {
  for (int off = width >> 1; off > 0; off >>= 1) {
    T ov = __shfl_xor(v, off, RAFT_AMD_WAVE);
    I oi = __shfl_xor(i, off, RAFT_AMD_WAVE);
    if (ov < v || (ov == v && oi < i)) { v = ov; i = oi; }
  }
}

// ---------------------------------------------------------------------------
// block-level reductions: wave reduce -> LDS tree across waves
// (reference parity: raft/util/reduction.cuh blockReduce)
// ---------------------------------------------------------------------------

template <typename T, int BLOCK>
__device__ __forceinline__ T block_reduce_sum(T v, T* lds /* BLOCK/64 */) {
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  v = wave_reduce_sum(v);
  const int wid = threadIdx.x / RAFT_AMD_WAVE;
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  if (wid == 0) {
    v = lane < NW ? lds[lane] : T(0);
    v = wave_reduce_sum(v, NW);
  }
  return v;  // valid in wave 0
}

template <typename T, typename I, int BLOCK>
__device__ __forceinline__ void block_reduce_argmin(T& v, I& i, T* lds_v, I* lds_i) {
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  wave_reduce_argmin(v, i);
  const int wid = threadIdx.x / RAFT_AMD_WAVE;
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  if (lane == 0) { lds_v[wid] = v; lds_i[wid] = i; }
  __syncthreads();
  if (wid == 0) {
    if (lane < NW) { v = lds_v[lane]; i = lds_i[lane]; }
    else { v = lds_v[0]; i = lds_i[0]; }
    wave_reduce_argmin(v, i);
  }
}

// ---------------------------------------------------------------------------
// Kahan-Babushka-Neumaier compensated accumulator (reference parity:
// coalesced_reduction-inl.cuh:36-45 — keeps fp32 row sums accurate at D>=2^17)
// ---------------------------------------------------------------------------

template <typename T>
struct KahanAcc {
  T sum = T(0);
  T c = T(0);
  __device__ __forceinline__ void add(T x) {
    T t = sum + x;
    if (fabsf((float)sum) >= fabsf((float)x)) c += (sum - t) + x;
    else c += (x - t) + sum;
    sum = t;
  }
  __device__ __forceinline__ T get() const { return sum + c; }
};

// grid sizing for memory-bound grid-stride kernels (guide G11):
// cap at ~8 blocks/CU on 256 CUs and stride the rest.
inline int grid_1d(long long total_threads, int block) {
  long long blocks = (total_threads + block - 1) / block;
  long long cap = 2048;
  return (int)(blocks < cap ? blocks : cap);
}

}  // namespace raft_amd
