// Row/column reductions, row argmin, fused row-normalize — wave64 CDNA4.
//
// Reference parity (WHAT): raft/linalg/detail/coalesced_reduction-inl.cuh
// (thin/medium regimes + Kahan adds), strided_reduction.cuh, matrix argmin,
// linalg detail/normalize.cuh. The kernel geometry here is CDNA4-native:
// logical warps are 2..64 lanes of a 64-wide wavefront; column reductions
// assign consecutive lanes to consecutive columns (coalesced along the row);
// all f32 loads on the row path are float4-vectorized (guide G13).

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

// op codes (keep in sync with raft_amd/linalg/reduce.py _EXT_CODES)
// 0: sum(x), 1: sum(x^2), 2: sum(|x|), 3: max(x), 4: min(x), 5: max(|x|)

template <int OP, typename T>
__device__ __forceinline__ T main_op(T x) {
  if constexpr (OP == 1) return x * x;
  if constexpr (OP == 2 || OP == 5) return x < T(0) ? -x : x;
  return x;
}

template <int OP>
constexpr bool is_sum() { return OP <= 2; }

template <int OP, typename T>
__device__ __forceinline__ T red_op(T a, T b) {
  if constexpr (OP <= 2) return a + b;
  else if constexpr (OP == 3 || OP == 5) return a > b ? a : b;
  else return a < b ? a : b;
}

template <int OP, typename T>
__device__ __forceinline__ T red_init() {
  if constexpr (OP <= 2) return T(0);
  else if constexpr (OP == 3 || OP == 5) return T(-INFINITY);
  else return T(INFINITY);
}

template <typename T>
__device__ __forceinline__ T wave_red_generic_sum(T v, int width) {
  for (int off = width >> 1; off > 0; off >>= 1) v += __shfl_xor(v, off, RAFT_AMD_WAVE);
  return v;
}

// --------------------------------------------------------------------------
// thin row-reduce: logical warp of LW lanes per row, LW in {2..64} by D.
// grid-stride over rows. Kahan for sums.
// --------------------------------------------------------------------------
template <int OP, int LW, typename T>
__global__ void reduce_rows_thin_kernel(const T* __restrict__ x, T* __restrict__ out,
                                        long long n_rows, long long d) {
  const long long lwarps_per_block = blockDim.x / LW;
  const long long lw_in_block = threadIdx.x / LW;
  const int lane = threadIdx.x % LW;
  long long row = (long long)blockIdx.x * lwarps_per_block + lw_in_block;
  const long long stride = (long long)gridDim.x * lwarps_per_block;
  for (; row < n_rows; row += stride) {
    const T* rp = x + row * d;
    if constexpr (is_sum<OP>()) {
      KahanAcc<T> acc;
      for (long long j = lane; j < d; j += LW) acc.add(main_op<OP>(rp[j]));
      T v = acc.get();
      for (int off = LW >> 1; off > 0; off >>= 1) v += __shfl_xor(v, off, RAFT_AMD_WAVE);
      if (lane == 0) out[row] = v;
    } else {
      T v = red_init<OP, T>();
      for (long long j = lane; j < d; j += LW) v = red_op<OP>(v, main_op<OP>(rp[j]));
      for (int off = LW >> 1; off > 0; off >>= 1)
        v = red_op<OP>(v, __shfl_xor(v, off, RAFT_AMD_WAVE));
      if (lane == 0) out[row] = v;
    }
  }
}

// --------------------------------------------------------------------------
// medium row-reduce: one 256-thread block per row, float4-vectorized loads.
// --------------------------------------------------------------------------
template <int OP, typename T, int BLOCK = 256>
__global__ void reduce_rows_block_kernel(const T* __restrict__ x, T* __restrict__ out,
                                         long long n_rows, long long d) {
  __shared__ T lds[BLOCK / RAFT_AMD_WAVE];
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* rp = x + row * d;
    T result;
    if constexpr (is_sum<OP>()) {
      KahanAcc<T> acc;
      if constexpr (sizeof(T) == 4) {
        const long long d4 = d / 4;
        const float4* rp4 = reinterpret_cast<const float4*>(rp);
        for (long long j = threadIdx.x; j < d4; j += BLOCK) {
          float4 v = rp4[j];
          acc.add(main_op<OP>((T)v.x)); acc.add(main_op<OP>((T)v.y));
          acc.add(main_op<OP>((T)v.z)); acc.add(main_op<OP>((T)v.w));
        }
        for (long long j = d4 * 4 + threadIdx.x; j < d; j += BLOCK)
          acc.add(main_op<OP>(rp[j]));
      } else {
        for (long long j = threadIdx.x; j < d; j += BLOCK) acc.add(main_op<OP>(rp[j]));
      }
      T v = acc.get();
      // block tree: wave sums then LDS
      v = wave_reduce_sum(v);
      const int wid = threadIdx.x / RAFT_AMD_WAVE, lane = threadIdx.x % RAFT_AMD_WAVE;
      if (lane == 0) lds[wid] = v;
      __syncthreads();
      if (threadIdx.x == 0) {
        result = T(0);
        for (int w = 0; w < BLOCK / RAFT_AMD_WAVE; w++) result += lds[w];
        out[row] = result;
      }
      __syncthreads();
    } else {
      T v = red_init<OP, T>();
      for (long long j = threadIdx.x; j < d; j += BLOCK) v = red_op<OP>(v, main_op<OP>(rp[j]));
      for (int off = RAFT_AMD_WAVE >> 1; off > 0; off >>= 1)
        v = red_op<OP>(v, __shfl_xor(v, off, RAFT_AMD_WAVE));
      const int wid = threadIdx.x / RAFT_AMD_WAVE, lane = threadIdx.x % RAFT_AMD_WAVE;
      if (lane == 0) lds[wid] = v;
      __syncthreads();
      if (threadIdx.x == 0) {
        result = lds[0];
        for (int w = 1; w < BLOCK / RAFT_AMD_WAVE; w++) result = red_op<OP>(result, lds[w]);
        out[row] = result;
      }
      __syncthreads();
    }
  }
}

// --------------------------------------------------------------------------
// column reduce: consecutive lanes own consecutive columns (coalesced);
// each block tiles BLOCK columns and walks all rows.
// --------------------------------------------------------------------------
// 2D grid: blockIdx.x tiles columns, blockIdx.y tiles rows — a skinny
// matrix (d << 256) still fills the chip. Each block reduces its row chunk
// (Kahan for sums) then combines across row tiles with device atomics
// (CAS loop for min/max since HIP has no float atomicMin/Max) — the same
// partial+atomic shape as the reference stridedSummationKernel.
template <int OP, typename T, int BLOCK = 256>
__global__ void reduce_cols_kernel(const T* __restrict__ x, T* __restrict__ out,
                                   long long n_rows, long long d,
                                   long long rows_per_tile) {
  const long long col = (long long)blockIdx.x * BLOCK + threadIdx.x;
  if (col >= d) return;
  const long long r0 = (long long)blockIdx.y * rows_per_tile;
  const long long r1 = min(n_rows, r0 + rows_per_tile);
  if constexpr (is_sum<OP>()) {
    // 4 independent Kahan accumulators: the compensated add is a serial
    // dependency chain, so a single accumulator caps each thread at
    // ~1 element / chain-latency — 4-way ILP restores bandwidth
    KahanAcc<T> a0, a1, a2, a3;
    long long r = r0;
    for (; r + 3 < r1; r += 4) {
      a0.add(main_op<OP>(x[r * d + col]));
      a1.add(main_op<OP>(x[(r + 1) * d + col]));
      a2.add(main_op<OP>(x[(r + 2) * d + col]));
      a3.add(main_op<OP>(x[(r + 3) * d + col]));
    }
    for (; r < r1; r++) a0.add(main_op<OP>(x[r * d + col]));
    a0.add(a1.get());
    a0.add(a2.get());
    a0.add(a3.get());
    if (gridDim.y == 1) {
      out[col] = a0.get();
    } else {
      atomicAdd(&out[col], a0.get());
    }
  } else {
    T v0 = red_init<OP, T>(), v1 = v0, v2 = v0, v3 = v0;
    long long r = r0;
    for (; r + 3 < r1; r += 4) {
      v0 = red_op<OP>(v0, main_op<OP>(x[r * d + col]));
      v1 = red_op<OP>(v1, main_op<OP>(x[(r + 1) * d + col]));
      v2 = red_op<OP>(v2, main_op<OP>(x[(r + 2) * d + col]));
      v3 = red_op<OP>(v3, main_op<OP>(x[(r + 3) * d + col]));
    }
    for (; r < r1; r++) v0 = red_op<OP>(v0, main_op<OP>(x[r * d + col]));
    T v = red_op<OP>(red_op<OP>(v0, v1), red_op<OP>(v2, v3));
    if (gridDim.y == 1) {
      out[col] = v;
    } else if constexpr (sizeof(T) == 4) {
      int* addr = reinterpret_cast<int*>(&out[col]);
      int cur = __float_as_int(*reinterpret_cast<volatile float*>(addr));
      while (true) {
        const float merged = (float)red_op<OP>((T)__int_as_float(cur), v);
        const int old = atomicCAS(addr, cur, __float_as_int(merged));
        if (old == cur) break;
        cur = old;
      }
    } else {
      unsigned long long* addr = reinterpret_cast<unsigned long long*>(&out[col]);
      unsigned long long cur = *reinterpret_cast<volatile unsigned long long*>(addr);
      while (true) {
        double merged = (double)red_op<OP>((T)__longlong_as_double(cur), v);
        const unsigned long long old =
            atomicCAS(addr, cur, __double_as_longlong(merged));
        if (old == cur) break;
        cur = old;
      }
    }
  }
}

template <typename T>
__global__ void fill_kernel(T* __restrict__ out, long long n, T v) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = v;
}

// skinny-d column reduce (d < 32): a flat col-per-thread map would idle
// 64-d lanes of every wave, so lanes cover (row_sub, col) instead —
// RPT=64/ceil_pow2(d) rows per wave step, full coalescing — and the row_sub
// partials fold with log2(RPT) shuffles before the cross-tile atomic.
template <int OP, typename T, int BLOCK = 256>
__global__ void reduce_cols_skinny_kernel(const T* __restrict__ x,
                                          T* __restrict__ out, long long n_rows,
                                          long long d, long long rows_per_tile) {
  int dp2 = 1;
  while (dp2 < d) dp2 <<= 1;          // <= 32
  const int rpt = RAFT_AMD_WAVE / dp2; // rows per wave step
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int wv = threadIdx.x / RAFT_AMD_WAVE;
  const int col = lane % dp2;
  const int rsub = lane / dp2;
  const long long r0 = (long long)blockIdx.x * rows_per_tile;
  const long long r1 = min(n_rows, r0 + rows_per_tile);
  const int waves = BLOCK / RAFT_AMD_WAVE;
  T v = red_init<OP, T>();
  KahanAcc<T> acc;
  if (col < d) {
    for (long long r = r0 + (long long)wv * rpt + rsub; r < r1;
         r += (long long)waves * rpt) {
      const T e = main_op<OP>(x[r * d + col]);
      if constexpr (is_sum<OP>()) acc.add(e);
      else v = red_op<OP>(v, e);
    }
  }
  // fold row_subs: shuffle down by dp2 strides within the wave
  if constexpr (is_sum<OP>()) v = acc.get();
  for (int off = RAFT_AMD_WAVE >> 1; off >= dp2; off >>= 1)
    v = red_op<OP>(v, __shfl_xor(v, off, RAFT_AMD_WAVE));
  // one lane per (wave, col) merges across waves + tiles via atomics
  if (rsub == 0 && col < d) {
    if constexpr (is_sum<OP>()) {
      atomicAdd(&out[col], v);
    } else if constexpr (sizeof(T) == 4) {
      int* addr = reinterpret_cast<int*>(&out[col]);
      int cur = __float_as_int(*reinterpret_cast<volatile float*>(addr));
      while (true) {
        const float merged = (float)red_op<OP>((T)__int_as_float(cur), v);
        const int old = atomicCAS(addr, cur, __float_as_int(merged));
        if (old == cur) break;
        cur = old;
      }
    } else {
      unsigned long long* addr = reinterpret_cast<unsigned long long*>(&out[col]);
      unsigned long long cur = *reinterpret_cast<volatile unsigned long long*>(addr);
      while (true) {
        double merged = (double)red_op<OP>((T)__longlong_as_double(cur), v);
        const unsigned long long old =
            atomicCAS(addr, cur, __double_as_longlong(merged));
        if (old == cur) break;
        cur = old;
      }
    }
  }
}

// --------------------------------------------------------------------------
// row argmin: one wave per row (thin) or block per row (wide)
// --------------------------------------------------------------------------
template <typename T, int BLOCK = 256>
__global__ void row_argmin_kernel(const T* __restrict__ x, int* __restrict__ out,
                                  long long n_rows, long long d) {
  __shared__ T lds_v[BLOCK / RAFT_AMD_WAVE];
  __shared__ int lds_i[BLOCK / RAFT_AMD_WAVE];
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* rp = x + row * d;
    T v = (T)INFINITY;
    int vi = 0;
    for (long long j = threadIdx.x; j < d; j += BLOCK) {
      T t = rp[j];
      if (t < v || (t == v && (long long)vi > j)) { v = t; vi = (int)j; }
    }
    block_reduce_argmin<T, int, BLOCK>(v, vi, lds_v, lds_i);
    if (threadIdx.x == 0) out[row] = vi;
    __syncthreads();
  }
}

// --------------------------------------------------------------------------
// fused L2 row-normalize: one block per row; sumsq in LDS-broadcast then scale
// (one extra read beats two kernel launches; row stays in L2 between passes)
// --------------------------------------------------------------------------
template <typename T, int BLOCK = 256>
__global__ void row_normalize_l2_kernel(const T* __restrict__ x, T* __restrict__ out,
                                        long long n_rows, long long d, T eps) {
  __shared__ T lds[BLOCK / RAFT_AMD_WAVE];
  __shared__ T inv_norm;
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T* rp = x + row * d;
    KahanAcc<T> acc;
    for (long long j = threadIdx.x; j < d; j += BLOCK) { T t = rp[j]; acc.add(t * t); }
    T v = wave_reduce_sum(acc.get());
    const int wid = threadIdx.x / RAFT_AMD_WAVE, lane = threadIdx.x % RAFT_AMD_WAVE;
    if (lane == 0) lds[wid] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      T s = T(0);
      for (int w = 0; w < BLOCK / RAFT_AMD_WAVE; w++) s += lds[w];
      T n = sqrtf((float)s);
      inv_norm = T(1) / (n > eps ? n : eps);
    }
    __syncthreads();
    const T inv = inv_norm;
    T* op = out + row * d;
    for (long long j = threadIdx.x; j < d; j += BLOCK) op[j] = rp[j] * inv;
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// host-side dispatch
// ---------------------------------------------------------------------------

template <int OP, typename T>
void launch_reduce_rows(const T* x, T* out, long long n_rows, long long d,
                        hipStream_t stream) {
  if (d <= 256) {
    // thin: logical warp width by D (reference heuristic re-derived for wave64)
    constexpr int BLOCK = 256;
    int grid;
    if (d <= 4) {
      grid = grid_1d(n_rows * 2, BLOCK);
      hipLaunchKernelGGL((reduce_rows_thin_kernel<OP, 2, T>), dim3(grid), dim3(BLOCK), 0, stream, x, out, n_rows, d);
    } else if (d <= 16) {
      grid = grid_1d(n_rows * 8, BLOCK);
      hipLaunchKernelGGL((reduce_rows_thin_kernel<OP, 8, T>), dim3(grid), dim3(BLOCK), 0, stream, x, out, n_rows, d);
    } else if (d <= 64) {
      grid = grid_1d(n_rows * 32, BLOCK);
      hipLaunchKernelGGL((reduce_rows_thin_kernel<OP, 32, T>), dim3(grid), dim3(BLOCK), 0, stream, x, out, n_rows, d);
    } else {
      grid = grid_1d(n_rows * 64, BLOCK);
      hipLaunchKernelGGL((reduce_rows_thin_kernel<OP, 64, T>), dim3(grid), dim3(BLOCK), 0, stream, x, out, n_rows, d);
    }
  } else {
    int grid = (int)(n_rows < 2048 ? n_rows : 2048);
    hipLaunchKernelGGL((reduce_rows_block_kernel<OP, T>), dim3(grid), dim3(256), 0, stream, x, out, n_rows, d);
  }
}

template <int OP, typename T>
void launch_reduce_cols(const T* x, T* out, long long n_rows, long long d,
                        hipStream_t stream) {
  if (d < 32 && n_rows >= 4096) {
    long long gy = min((long long)1024, (n_rows + 4095) / 4096);
    const long long rows_per_tile = (n_rows + gy - 1) / gy;
    if constexpr (OP <= 2) {
      hipMemsetAsync(out, 0, d * sizeof(T), stream);
    } else {
      const T init = (OP == 3 || OP == 5) ? (T)-INFINITY : (T)INFINITY;
      hipLaunchKernelGGL((fill_kernel<T>), dim3(1), dim3(256), 0, stream, out,
                         d, init);
    }
    hipLaunchKernelGGL((reduce_cols_skinny_kernel<OP, T>), dim3((int)gy),
                       dim3(256), 0, stream, x, out, n_rows, d, rows_per_tile);
    return;
  }
  const int gx = (int)((d + 255) / 256);
  // enough row tiles to fill 256 CUs x 2 blocks even for skinny d, but keep
  // each chunk >= 1024 rows so atomic traffic stays negligible
  long long gy = (512 + gx - 1) / gx;
  gy = min(gy, (n_rows + 1023) / 1024);
  if (gy < 1) gy = 1;
  const long long rows_per_tile = (n_rows + gy - 1) / gy;
  if (gy > 1) {
    if constexpr (OP <= 2) {
      hipMemsetAsync(out, 0, d * sizeof(T), stream);
    } else {
      const T init = (OP == 3 || OP == 5) ? (T)-INFINITY : (T)INFINITY;
      hipLaunchKernelGGL((fill_kernel<T>), dim3((int)((d + 255) / 256)),
                         dim3(256), 0, stream, out, d, init);
    }
  }
  hipLaunchKernelGGL((reduce_cols_kernel<OP, T>), dim3(gx, (int)gy), dim3(256),
                     0, stream, x, out, n_rows, d, rows_per_tile);
}

#define INSTANTIATE_OPS(T)                                                              \
  template void launch_reduce_rows<0, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_rows<1, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_rows<2, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_rows<3, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_rows<4, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_rows<5, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_cols<0, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_cols<1, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_cols<2, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_cols<3, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_cols<4, T>(const T*, T*, long long, long long, hipStream_t); \
  template void launch_reduce_cols<5, T>(const T*, T*, long long, long long, hipStream_t);

INSTANTIATE_OPS(float)
INSTANTIATE_OPS(double)

// bf16 row squared-norms -> f32 (vectorized bf16x8 loads; avoids the 2x-size
// fp32 materialization a torch x.float().pow(2).sum(1) chain would need)
__global__ void rows_sqnorm_bf16_kernel(const __bf16* __restrict__ x,
                                        float* __restrict__ out,
                                        long long n_rows, long long d) {
  typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
  constexpr int BLOCK = 256;
  __shared__ float lds[BLOCK / RAFT_AMD_WAVE];
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const __bf16* rp = x + row * d;
    float acc = 0.f;
    const long long d8 = d / 8;
    const bf16x8_t* rp8 = reinterpret_cast<const bf16x8_t*>(rp);
    for (long long j = threadIdx.x; j < d8; j += BLOCK) {
      const bf16x8_t v = rp8[j];
#pragma unroll
      for (int e = 0; e < 8; e++) {
        const float f = (float)v[e];
        acc += f * f;
      }
    }
    for (long long j = d8 * 8 + threadIdx.x; j < d; j += BLOCK) {
      const float f = (float)rp[j];
      acc += f * f;
    }
    acc = wave_reduce_sum(acc);
    const int wid = threadIdx.x / RAFT_AMD_WAVE, lane = threadIdx.x % RAFT_AMD_WAVE;
    if (lane == 0) lds[wid] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      float s = 0.f;
      for (int w = 0; w < BLOCK / RAFT_AMD_WAVE; w++) s += lds[w];
      out[row] = s;
    }
    __syncthreads();
  }
}

void launch_rows_sqnorm_bf16(const void* x, float* out, long long n_rows,
                             long long d, hipStream_t stream) {
  long long blocks = n_rows < 65536 ? n_rows : 65536;
  hipLaunchKernelGGL(rows_sqnorm_bf16_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     stream, (const __bf16*)x, out, n_rows, d);
}

void launch_row_argmin(const float* x, int* out, long long n_rows, long long d,
                       hipStream_t stream) {
  int grid = (int)(n_rows < 2048 ? n_rows : 2048);
  hipLaunchKernelGGL((row_argmin_kernel<float>), dim3(grid), dim3(256), 0, stream, x, out, n_rows, d);
}

void launch_row_normalize_l2(const float* x, float* out, long long n_rows, long long d,
                             float eps, hipStream_t stream) {
  int grid = (int)(n_rows < 2048 ? n_rows : 2048);
  hipLaunchKernelGGL((row_normalize_l2_kernel<float>), dim3(grid), dim3(256), 0, stream, x, out, n_rows, d, eps);
}

}  // namespace raft_amd
