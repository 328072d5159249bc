// Batched top-k selection — block-per-row MSB radix select (wave64 CDNA4).
//
// Reference parity (WHAT): raft/matrix/detail/select_radix.cuh (multi-pass
// MSB-first radix with histogram + bucket choose, candidate compaction, and
// the one-block-per-row variant radix_topk_one_block_kernel:1040) and the
// optional post-sort.
//
// MI355X design:
//  * one block per row, float4-vectorized scans, per-wave LDS histograms
//    (merged after) to avoid LDS-atomic serialization;
//  * 2 full-row passes: (1) MSB histogram, (2) emit sure winners + compact
//    the boundary bucket into a per-block GLOBAL workspace (keeping LDS at
//    ~21 KiB for multi-block occupancy — an LDS candidate buffer measured
//    1 block/CU and 5x slower); remaining radix passes refine inside the
//    compacted candidates (expected len/256 elements, L2-resident);
//  * skewed rows overflowing the workspace fall back to full-row passes;
//  * final top-k pairs bitonic-sorted in LDS (k <= 2048).

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

__device__ __forceinline__ uint32_t f32_to_ord(float f, bool select_min) {
  // Canonicalize NaN (any payload, either sign) to the maximum ordinal so it
  // orders AFTER every finite value and +/-inf in BOTH selection directions.
  // Without this a negative-sign NaN maps below -inf and is selected first.
  if (f != f) return 0xFFFFFFFFu;
  uint32_t u = __float_as_uint(f);
  u = (u & 0x80000000u) ? ~u : (u | 0x80000000u);  // monotone: asc float -> asc u
  return select_min ? u : ~u;
}

__device__ __forceinline__ float ord_to_f32(uint32_t u, bool select_min) {
  if (!select_min) u = ~u;
  u = (u & 0x80000000u) ? (u & 0x7FFFFFFFu) : ~u;
  return __uint_as_float(u);
}

constexpr int SELECT_K_MAX = 2048;
constexpr int CAND_CAP = 16384;  // per-block gmem candidate slots

// sort (optional) + write the k selected pairs for one row
template <int BLOCK>
__device__ __forceinline__ void select_k_finish(const float* __restrict__ rp,
                                                float* __restrict__ out_v,
                                                int* __restrict__ out_i,
                                                long long row, int k, bool do_sort,
                                                uint32_t* pair_u, int* pair_i) {
  if (do_sort) {
    int kp = 1;
    while (kp < k) kp <<= 1;
    // padding must tie-break AFTER real entries (canonical NaN shares the
    // max ordinal), so the index sentinel is INT_MAX, not -1
    for (int j = threadIdx.x + k; j < kp; j += BLOCK) {
      if (j < SELECT_K_MAX) { pair_u[j] = 0xFFFFFFFFu; pair_i[j] = INT_MAX; }
    }
    __syncthreads();
    for (int size = 2; size <= kp; size <<= 1) {
      for (int strd = size >> 1; strd > 0; strd >>= 1) {
        for (int t = threadIdx.x; t < kp / 2; t += BLOCK) {
          const int i0 = 2 * t - (t & (strd - 1));
          const int i1 = i0 + strd;
          const bool up = ((i0 & size) == 0);
          const uint32_t a = pair_u[i0], b = pair_u[i1];
          const bool swap_ = (a > b || (a == b && pair_i[i0] > pair_i[i1])) == up;
          if (swap_) {
            pair_u[i0] = b; pair_u[i1] = a;
            const int ti = pair_i[i0]; pair_i[i0] = pair_i[i1]; pair_i[i1] = ti;
          }
        }
        __syncthreads();
      }
    }
  }
  for (int j = threadIdx.x; j < k; j += BLOCK) {
    const int src = pair_i[j];
    out_v[row * k + j] = rp[src];
    out_i[row * k + j] = src;
  }
  __syncthreads();
}

// per-wave histogram helper: each wave owns hist[wave][256]; merged by caller
template <int BLOCK>
__device__ __forceinline__ void merge_hists(unsigned int (*hist)[256]) {
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  __syncthreads();
  for (int b = threadIdx.x; b < 256; b += BLOCK) {
    unsigned int s = 0;
#pragma unroll
    for (int w = 0; w < NW; w++) s += hist[w][b];
    hist[0][b] = s;
  }
  __syncthreads();
}

template <int BLOCK = 256>
__global__ void select_k_radix_kernel(const float* __restrict__ x,
                                      float* __restrict__ out_v,
                                      int* __restrict__ out_i,
                                      uint2* __restrict__ cand_ws,
                                      long long batch, long long len, int k,
                                      bool select_min, bool do_sort) {
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  __shared__ unsigned int hist[NW][256];
  __shared__ unsigned int sh_prefix, sh_below, sh_cnt_lt, sh_cnt_eq, sh_cand_cnt;
  __shared__ uint32_t pair_u[SELECT_K_MAX];
  __shared__ int pair_i[SELECT_K_MAX];
  uint2* cand = cand_ws + (long long)blockIdx.x * CAND_CAP;
  const int wid = threadIdx.x / RAFT_AMD_WAVE;

  for (long long row = blockIdx.x; row < batch; row += gridDim.x) {
    const float* rp = x + row * len;
    const long long len4 = len / 4;
    const float4* rp4 = reinterpret_cast<const float4*>(rp);

    // ---- pass 1: MSB histogram (vectorized full scan, per-wave hists) ----
    for (int b = threadIdx.x; b < NW * 256; b += BLOCK)
      reinterpret_cast<unsigned int*>(hist)[b] = 0;
    if (threadIdx.x == 0) { sh_cnt_lt = 0; sh_cand_cnt = 0; }
    __syncthreads();
    for (long long j = threadIdx.x; j < len4; j += BLOCK) {
      const float4 v = rp4[j];
      atomicAdd(&hist[wid][f32_to_ord(v.x, select_min) >> 24], 1u);
      atomicAdd(&hist[wid][f32_to_ord(v.y, select_min) >> 24], 1u);
      atomicAdd(&hist[wid][f32_to_ord(v.z, select_min) >> 24], 1u);
      atomicAdd(&hist[wid][f32_to_ord(v.w, select_min) >> 24], 1u);
    }
    for (long long j = len4 * 4 + threadIdx.x; j < len; j += BLOCK)
      atomicAdd(&hist[wid][f32_to_ord(rp[j], select_min) >> 24], 1u);
    merge_hists<BLOCK>(hist);
    if (threadIdx.x == 0) {
      unsigned int cum = 0;
      int bucket = 255;
      for (int b = 0; b < 256; b++) {
        const unsigned int c = hist[0][b];
        if (cum + c >= (unsigned int)k) { bucket = b; break; }
        cum += c;
      }
      sh_prefix = (unsigned int)bucket;
      sh_below = cum;
    }
    __syncthreads();
    const uint32_t b1 = sh_prefix;
    const unsigned int below1 = sh_below;
    const int remaining1 = k - (int)below1;

    // ---- pass 2: emit sure winners (LDS), compact boundary bucket (gmem) --
    auto classify = [&](uint32_t u, long long j) {
      const uint32_t byte0 = u >> 24;
      if (byte0 < b1) {
        const unsigned int slot = atomicAdd(&sh_cnt_lt, 1u);
        pair_u[slot] = u;
        pair_i[slot] = (int)j;
      } else if (byte0 == b1) {
        const unsigned int c = atomicAdd(&sh_cand_cnt, 1u);
        if (c < CAND_CAP) cand[c] = uint2{u, (unsigned int)j};
      }
    };
    for (long long j = threadIdx.x; j < len4; j += BLOCK) {
      const float4 v = rp4[j];
      classify(f32_to_ord(v.x, select_min), j * 4 + 0);
      classify(f32_to_ord(v.y, select_min), j * 4 + 1);
      classify(f32_to_ord(v.z, select_min), j * 4 + 2);
      classify(f32_to_ord(v.w, select_min), j * 4 + 3);
    }
    for (long long j = len4 * 4 + threadIdx.x; j < len; j += BLOCK)
      classify(f32_to_ord(rp[j], select_min), j);
    __syncthreads();
    const unsigned int cand_cnt = sh_cand_cnt;

    uint32_t prefix = b1 << 24;
    uint32_t prefix_mask = 0xFFu << 24;
    int remaining = remaining1;

    if (cand_cnt <= CAND_CAP) {
      // ---- fast path: refine + collect inside the compacted candidates ---
      for (int pass = 1; pass < 4; pass++) {
        const int shift = 8 * (3 - pass);
        for (int b = threadIdx.x; b < NW * 256; b += BLOCK)
          reinterpret_cast<unsigned int*>(hist)[b] = 0;
        __syncthreads();
        for (unsigned int j = threadIdx.x; j < cand_cnt; j += BLOCK) {
          const uint32_t u = cand[j].x;
          if ((u & prefix_mask) == prefix)
            atomicAdd(&hist[wid][(u >> shift) & 0xFF], 1u);
        }
        merge_hists<BLOCK>(hist);
        if (threadIdx.x == 0) {
          unsigned int cum = 0;
          int bucket = 255;
          for (int b = 0; b < 256; b++) {
            const unsigned int c = hist[0][b];
            if (cum + c >= (unsigned int)remaining) { bucket = b; break; }
            cum += c;
          }
          sh_prefix = (unsigned int)bucket;
          sh_below = cum;
        }
        __syncthreads();
        remaining -= (int)sh_below;
        prefix |= (sh_prefix << shift);
        prefix_mask |= (0xFFu << shift);
        __syncthreads();
      }
      const uint32_t kth = prefix;
      const unsigned int n_lt2 = (unsigned int)(remaining1 - remaining);
      if (threadIdx.x == 0) { sh_cnt_lt = 0; sh_cnt_eq = 0; }
      __syncthreads();
      for (unsigned int j = threadIdx.x; j < cand_cnt; j += BLOCK) {
        const uint32_t u = cand[j].x;
        if (u < kth) {
          const unsigned int slot = atomicAdd(&sh_cnt_lt, 1u);
          pair_u[below1 + slot] = u;
          pair_i[below1 + slot] = (int)cand[j].y;
        } else if (u == kth) {
          const unsigned int e = atomicAdd(&sh_cnt_eq, 1u);
          if (e < (unsigned int)remaining) {
            pair_u[below1 + n_lt2 + e] = u;
            pair_i[below1 + n_lt2 + e] = (int)cand[j].y;
          }
        }
      }
      __syncthreads();
      select_k_finish<BLOCK>(rp, out_v, out_i, row, k, do_sort, pair_u, pair_i);
      continue;
    }

    // ---- fallback (boundary bucket overflowed the workspace): classic
    // multi-pass refinement + collection over the full row -----------------
    for (int pass = 1; pass < 4; pass++) {
      const int shift = 8 * (3 - pass);
      for (int b = threadIdx.x; b < NW * 256; b += BLOCK)
        reinterpret_cast<unsigned int*>(hist)[b] = 0;
      __syncthreads();
      for (long long j = threadIdx.x; j < len; j += BLOCK) {
        const uint32_t u = f32_to_ord(rp[j], select_min);
        if ((u & prefix_mask) == prefix)
          atomicAdd(&hist[wid][(u >> shift) & 0xFF], 1u);
      }
      merge_hists<BLOCK>(hist);
      if (threadIdx.x == 0) {
        unsigned int cum = 0;
        int bucket = 255;
        for (int b = 0; b < 256; b++) {
          const unsigned int c = hist[0][b];
          if (cum + c >= (unsigned int)remaining) { bucket = b; break; }
          cum += c;
        }
        sh_prefix = (unsigned int)bucket;
        sh_below = cum;
      }
      __syncthreads();
      remaining -= (int)sh_below;
      prefix |= (sh_prefix << shift);
      prefix_mask |= (0xFFu << shift);
      __syncthreads();
    }
    const uint32_t kth = prefix;
    if (threadIdx.x == 0) { sh_cnt_lt = 0; sh_cnt_eq = 0; }
    __syncthreads();
    const unsigned int n_lt = (unsigned int)(k - remaining);
    for (long long j = threadIdx.x; j < len; j += BLOCK) {
      const uint32_t u = f32_to_ord(rp[j], select_min);
      if (u < kth) {
        const unsigned int slot = atomicAdd(&sh_cnt_lt, 1u);
        pair_u[slot] = u;
        pair_i[slot] = (int)j;
      } else if (u == kth) {
        const unsigned int e = atomicAdd(&sh_cnt_eq, 1u);
        if (e < (unsigned int)remaining) {
          pair_u[n_lt + e] = u;
          pair_i[n_lt + e] = (int)j;
        }
      }
    }
    __syncthreads();
    select_k_finish<BLOCK>(rp, out_v, out_i, row, k, do_sort, pair_u, pair_i);
  }
}

// ---------------------------------------------------------------------------
// warpsort select (k <= 64): one wave per row, per-lane single (val,idx)
// queue slot — the wave-wide register priority queue of the reference's
// select_warpsort.cuh, re-derived for 64-lane wavefronts (capacity == wave
// width == 64, so one bitonic network IS the queue). Strategy =
// filtered-immediate: batches of 64 loads; a ballot skips batches with no
// candidate below the current worst (after warm-up nearly all batches skip,
// leaving a pure streaming read); candidate batches are wave-bitonic-sorted
// descending and merged with the ascending queue by elementwise min + one
// bitonic merge.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void cmp_exchange(uint32_t& v, int& i, int stride, bool keep_small) {
  const uint32_t ov = __shfl_xor(v, stride, RAFT_AMD_WAVE);
  const int oi = __shfl_xor(i, stride, RAFT_AMD_WAVE);
  const bool other_smaller = (ov < v) || (ov == v && oi < i);
  if (keep_small == other_smaller) { v = ov; i = oi; }
}

__device__ __forceinline__ void wave_bitonic_sort_asc(uint32_t& v, int& i, int lane) {
#pragma unroll
  for (int size = 2; size <= 64; size <<= 1) {
#pragma unroll
    for (int stride = 64 >> 1; stride > 0; stride >>= 1) {
      if (stride < size) {
        const bool up = (lane & size) == 0 || size == 64;
        const bool keep_small = ((lane & stride) == 0) == up;
        cmp_exchange(v, i, stride, keep_small);
      }
    }
  }
}

__device__ __forceinline__ void wave_bitonic_merge_asc(uint32_t& v, int& i, int lane) {
#pragma unroll
  for (int stride = 32; stride > 0; stride >>= 1) {
    const bool keep_small = (lane & stride) == 0;
    cmp_exchange(v, i, stride, keep_small);
  }
}

// The queue runs in the monotone ORDINAL domain (f32_to_ord), so NaN values
// participate with their real indices (canonical max ordinal = ordered last)
// and every output slot carries a valid in-range index — the float-domain
// version dropped NaN/inf ties and could leave -1 sentinels in the output.
template <int BLOCK = 256>
__global__ void select_k_warpsort_kernel(const float* __restrict__ x,
                                         float* __restrict__ out_v,
                                         int* __restrict__ out_i,
                                         long long batch, long long len, int k,
                                         bool select_min) {
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const long long waves_per_block = BLOCK / RAFT_AMD_WAVE;
  long long row = (long long)blockIdx.x * waves_per_block + threadIdx.x / RAFT_AMD_WAVE;
  const long long stride = (long long)gridDim.x * waves_per_block;
  for (; row < batch; row += stride) {
    const float* rp = x + row * len;
    uint32_t qv = 0xFFFFFFFFu;  // queue ascending across lanes; lane 63 = worst
    int qi = INT_MAX;           // index tie-break prefers real (smaller) indices
    uint32_t worst = 0xFFFFFFFFu;
    for (long long j0 = 0; j0 < len; j0 += RAFT_AMD_WAVE) {
      const long long j = j0 + lane;
      const uint32_t val =
          j < len ? f32_to_ord(rp[j], select_min) : 0xFFFFFFFFu;
      // admit strictly-better candidates; while the worst slot is still at the
      // max ordinal (unfilled or NaN) also admit max-ordinal elements so their
      // real indices displace the INT_MAX sentinels (index tie-break wins)
      const bool cand =
          (val < worst) || (val == worst && worst == 0xFFFFFFFFu && j < len);
      if (__ballot(cand) == 0ull) continue;
      // sort batch ascending, then reverse to descending via lane mirror
      uint32_t bv = cand ? val : 0xFFFFFFFFu;
      int bi = cand ? (int)j : INT_MAX;
      wave_bitonic_sort_asc(bv, bi, lane);
      const uint32_t rv = __shfl(bv, 63 - lane, RAFT_AMD_WAVE);
      const int ri = __shfl(bi, 63 - lane, RAFT_AMD_WAVE);
      // elementwise min of (asc queue, desc batch) -> bitonic; re-merge
      const bool take = (rv < qv) || (rv == qv && ri < qi);
      if (take) { qv = rv; qi = ri; }
      wave_bitonic_merge_asc(qv, qi, lane);
      worst = __shfl(qv, (k <= 64 ? k : 64) - 1, RAFT_AMD_WAVE);
    }
    if (lane < k) {
      out_v[row * k + lane] = qi == INT_MAX ? ord_to_f32(qv, select_min)
                                            : rp[qi];
      out_i[row * k + lane] = qi == INT_MAX ? 0 : qi;
    }
  }
}

void launch_select_k_warpsort(const float* x, float* out_v, int* out_i,
                              long long batch, long long len, int k,
                              bool select_min, hipStream_t stream) {
  const long long blocks = (batch + 3) / 4;
  const int grid = (int)(blocks < 65536 ? blocks : 65536);
  hipLaunchKernelGGL((select_k_warpsort_kernel<256>), dim3(grid), dim3(256), 0,
                     stream, x, out_v, out_i, batch, len, k, select_min);
}

// ---------------------------------------------------------------------------
// Generic select-k: any value dtype (fp32/bf16/fp16 via 32-bit ordinals,
// fp64 via 64-bit ordinals), UNBOUNDED k, int64 indices (rows may exceed
// 2^31 elements) and per-row VARIABLE lengths — the CSR adapter that does
// NOT densify. Block-per-row threshold+filter:
//   1. byte-wise ordinal histogram passes narrow to the k-th ordinal
//      (4 passes for 32-bit ordinals, 8 for fp64) — no candidate storage,
//      so k is unbounded;
//   2. two filter passes write ordinals < kth, then the equality backfill.
// Rows shorter than k pad value with +inf (select_min; -inf otherwise) and
// index -1 (DOCUMENTED sentinel of the variable-length path). Output is
// UNSORTED; callers sort the [batch, k] slab when requested (k can exceed
// any LDS sort capacity).
// Reference parity: matrix/detail/select_radix.cuh per-row len_i
// (:722-725) + sparse/matrix/detail/select_k-inl.cuh CSR layout adapter
// (:63-96); covers the MATRIX_SELECT_LARGE_TEST len>2^31 case via int64
// indices and long long scans.
// ---------------------------------------------------------------------------

template <typename T>
struct SelOrd;
template <>
struct SelOrd<float> {
  using ord = uint32_t;
  static __device__ __forceinline__ uint32_t to_ord(float f, bool mn) {
    return f32_to_ord(f, mn);
  }
};
template <>
struct SelOrd<__bf16> {
  using ord = uint32_t;
  static __device__ __forceinline__ uint32_t to_ord(__bf16 h, bool mn) {
    return f32_to_ord((float)h, mn);
  }
};
template <>
struct SelOrd<_Float16> {
  using ord = uint32_t;
  static __device__ __forceinline__ uint32_t to_ord(_Float16 h, bool mn) {
    return f32_to_ord((float)h, mn);
  }
};
template <>
struct SelOrd<double> {
  using ord = uint64_t;
  static __device__ __forceinline__ uint64_t to_ord(double f, bool mn) {
    if (f != f) return ~0ull;
    uint64_t u = __double_as_longlong(f);
    u = (u & 0x8000000000000000ull) ? ~u : (u | 0x8000000000000000ull);
    return mn ? u : ~u;
  }
};

template <typename T, int BLOCK = 256>
__global__ void select_k_generic_kernel(const T* __restrict__ vals,
                                        const long long* __restrict__ row_off,
                                        long long stride, long long len_fixed,
                                        T* __restrict__ out_v,
                                        long long* __restrict__ out_i,
                                        long long batch, long long k,
                                        bool select_min) {
  using OrdT = typename SelOrd<T>::ord;
  constexpr int PASSES = (int)sizeof(OrdT);
  constexpr int NW = BLOCK / RAFT_AMD_WAVE;
  __shared__ unsigned long long hist[NW][256];
  __shared__ unsigned long long sh_prefix, sh_below, sh_lt, sh_eq;
  const int wid = threadIdx.x / RAFT_AMD_WAVE;

  for (long long row = blockIdx.x; row < batch; row += gridDim.x) {
    const long long base = row_off ? row_off[row] : row * stride;
    const long long len = row_off ? row_off[row + 1] - row_off[row] : len_fixed;
    const T* rp = vals + base;
    const long long kk = k < len ? k : len;
    if (kk <= 0) {
      for (long long j = threadIdx.x; j < k; j += BLOCK) {
        out_v[row * k + j] = (T)(select_min ? INFINITY : -INFINITY);
        out_i[row * k + j] = -1;
      }
      continue;
    }
    OrdT prefix = 0, prefix_mask = 0;
    long long remaining = kk;
#pragma unroll
    for (int pass = 0; pass < PASSES; pass++) {
      const int shift = 8 * (PASSES - 1 - pass);
      for (int b = threadIdx.x; b < NW * 256; b += BLOCK)
        reinterpret_cast<unsigned long long*>(hist)[b] = 0;
      __syncthreads();
      for (long long j = threadIdx.x; j < len; j += BLOCK) {
        const OrdT u = SelOrd<T>::to_ord(rp[j], select_min);
        if ((u & prefix_mask) == prefix)
          atomicAdd(&hist[wid][(u >> shift) & 0xFF], 1ull);
      }
      __syncthreads();
      for (int b = threadIdx.x; b < 256; b += BLOCK) {
        unsigned long long s = 0;
#pragma unroll
        for (int w = 0; w < NW; w++) s += hist[w][b];
        hist[0][b] = s;
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        unsigned long long cum = 0;
        int bucket = 255;
        for (int b = 0; b < 256; b++) {
          const unsigned long long c = hist[0][b];
          if (cum + c >= (unsigned long long)remaining) { bucket = b; break; }
          cum += c;
        }
        sh_prefix = (unsigned long long)bucket;
        sh_below = cum;
      }
      __syncthreads();
      prefix |= ((OrdT)sh_prefix) << shift;
      prefix_mask |= ((OrdT)0xFF) << shift;
      remaining -= (long long)sh_below;
      __syncthreads();
    }
    const OrdT kth = prefix;
    if (threadIdx.x == 0) { sh_lt = 0; sh_eq = 0; }
    __syncthreads();
    // filter pass 1: strict winners
    for (long long j = threadIdx.x; j < len; j += BLOCK) {
      const OrdT u = SelOrd<T>::to_ord(rp[j], select_min);
      if (u < kth) {
        const unsigned long long pos = atomicAdd(&sh_lt, 1ull);
        out_v[row * k + (long long)pos] = rp[j];
        out_i[row * k + (long long)pos] = j;
      }
    }
    __syncthreads();
    const long long n_lt = (long long)sh_lt;
    // filter pass 2: equality backfill (exactly kk - n_lt slots)
    for (long long j = threadIdx.x; j < len; j += BLOCK) {
      const OrdT u = SelOrd<T>::to_ord(rp[j], select_min);
      if (u == kth) {
        const unsigned long long e = atomicAdd(&sh_eq, 1ull);
        if ((long long)e < kk - n_lt) {
          out_v[row * k + n_lt + (long long)e] = rp[j];
          out_i[row * k + n_lt + (long long)e] = j;
        }
      }
    }
    // pad short rows
    for (long long j = kk + threadIdx.x; j < k; j += BLOCK) {
      out_v[row * k + j] = (T)(select_min ? INFINITY : -INFINITY);
      out_i[row * k + j] = -1;
    }
    __syncthreads();
  }
}

template <typename T>
void launch_select_k_generic_t(const T* vals, const long long* row_off,
                               long long stride, long long len_fixed, T* out_v,
                               long long* out_i, long long batch, long long k,
                               bool select_min, hipStream_t stream) {
  const int grid = (int)(batch < 4096 ? batch : 4096);
  hipLaunchKernelGGL((select_k_generic_kernel<T, 256>), dim3(grid), dim3(256),
                     0, stream, vals, row_off, stride, len_fixed, out_v, out_i,
                     batch, k, select_min);
}

template void launch_select_k_generic_t<float>(const float*, const long long*,
                                               long long, long long, float*,
                                               long long*, long long, long long,
                                               bool, hipStream_t);
template void launch_select_k_generic_t<double>(const double*, const long long*,
                                                long long, long long, double*,
                                                long long*, long long,
                                                long long, bool, hipStream_t);
template void launch_select_k_generic_t<__bf16>(const __bf16*, const long long*,
                                                long long, long long, __bf16*,
                                                long long*, long long,
                                                long long, bool, hipStream_t);
template void launch_select_k_generic_t<_Float16>(const _Float16*,
                                                  const long long*, long long,
                                                  long long, _Float16*,
                                                  long long*, long long,
                                                  long long, bool, hipStream_t);

int select_k_grid(long long batch) { return (int)(batch < 4096 ? batch : 4096); }

long long select_k_workspace_bytes(long long batch) {
  return (long long)select_k_grid(batch) * CAND_CAP * sizeof(uint2);
}

void launch_select_k(const float* x, float* out_v, int* out_i, void* cand_ws,
                     long long batch, long long len, int k, bool select_min,
                     bool do_sort, hipStream_t stream) {
  if (k > SELECT_K_MAX) throw std::runtime_error("select_k native path supports k <= 2048");
  const int grid = select_k_grid(batch);
  hipLaunchKernelGGL((select_k_radix_kernel<256>), dim3(grid), dim3(256), 0, stream,
                     x, out_v, out_i, (uint2*)cand_ws, batch, len, k, select_min,
                     do_sort);
}

}  // namespace raft_amd
