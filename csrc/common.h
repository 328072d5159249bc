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

// ---------------------------------------------------------------------------
// integer utilities (reference parity: raft/util/pow2_utils.cuh,
// fast_int_div.cuh, integer_utils.hpp)
// ---------------------------------------------------------------------------

template <typename T>
__host__ __device__ __forceinline__ constexpr T ceildiv(T a, T b) {
  return (a + b - 1) / b;
}

template <typename T>
__host__ __device__ __forceinline__ constexpr T round_up(T a, T b) {
  return ceildiv(a, b) * b;
}

// compile-time power-of-two helper: masks/shifts instead of div/mod
template <long long V>
struct Pow2 {
  static_assert((V & (V - 1)) == 0 && V > 0, "Pow2: V must be a power of 2");
  static constexpr long long value = V;
  static constexpr long long mask = V - 1;
  static constexpr int log2 = (V == 1) ? 0 : 1 + Pow2<V / 2>::log2;
  template <typename T>
  __host__ __device__ __forceinline__ static constexpr T div(T x) {
    return x >> log2;
  }
  template <typename T>
  __host__ __device__ __forceinline__ static constexpr T mod(T x) {
    return x & (T)mask;
  }
  template <typename T>
  __host__ __device__ __forceinline__ static constexpr T round_up_(T x) {
    return (x + (T)mask) & ~(T)mask;
  }
};
template <>
struct Pow2<1> {
  static constexpr long long value = 1, mask = 0;
  static constexpr int log2 = 0;
  template <typename T>
  __host__ __device__ __forceinline__ static constexpr T div(T x) { return x; }
  template <typename T>
  __host__ __device__ __forceinline__ static constexpr T mod(T) { return 0; }
  template <typename T>
  __host__ __device__ __forceinline__ static constexpr T round_up_(T x) {
    return x;
  }
};

// runtime fast division by an invariant divisor (magic-number method,
// Granlund-Montgomery): one 32x32 mulhi + shift instead of v_div
struct FastIntDiv {
  uint32_t d, magic;
  int shift;
  __host__ __device__ explicit FastIntDiv(uint32_t divisor) : d(divisor) {
    shift = 0;
    uint32_t t = divisor - 1;
    while (t >>= 1) shift++;
    shift += 1;  // ceil(log2(d))
    if ((divisor & (divisor - 1)) == 0) {
      magic = 0;  // power of two: pure shift
      shift = 0;
      uint32_t v = divisor;
      while (v >>= 1) shift++;
    } else {
      const uint64_t m = ((1ull << (32 + shift)) + divisor - 1) / divisor;
      magic = (uint32_t)m;
    }
  }
  __host__ __device__ __forceinline__ uint32_t div(uint32_t n) const {
    if (magic == 0) return n >> shift;
    return (uint32_t)(((uint64_t)n * magic) >> 32) >> shift;
  }
  __host__ __device__ __forceinline__ uint32_t mod(uint32_t n) const {
    return n - div(n) * d;
  }
};

// ---------------------------------------------------------------------------
// float atomic max/min via int-ordered CAS-free atomics (reference parity:
// raft/util/device_atomics.cuh; the minmax.cuh encode_traits trick) —
// NON-NEGATIVE floats only for the int-ordered fast path; general values
// use the ordered-uint encoding.
// ---------------------------------------------------------------------------

// classic sign-split trick on PLAIN float storage: non-negative IEEE floats
// order as signed ints, negative ones reverse-order as unsigned ints
__device__ __forceinline__ void atomic_max_float(float* p, float v) {
  if (v >= 0.f)
    atomicMax(reinterpret_cast<int*>(p), __float_as_int(v));
  else
    atomicMin(reinterpret_cast<unsigned int*>(p), __float_as_uint(v));
}
__device__ __forceinline__ void atomic_min_float(float* p, float v) {
  if (v >= 0.f)
    atomicMin(reinterpret_cast<int*>(p), __float_as_int(v));
  else
    atomicMax(reinterpret_cast<unsigned int*>(p), __float_as_uint(v));
}

}  // namespace raft_amd
