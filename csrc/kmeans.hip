// k-means centroid update: reduce_rows_by_key with replicated accumulators.
//
// Reference parity (WHAT): raft/linalg/reduce_rows_by_key (detail, smem-binned
// + atomic kernels). MI355X design: one wave per input row; lanes stride the
// feature dim with float4 vector loads and issue device-scope fp32 atomicAdds.
// Naive single-buffer atomics measured 7.2 ms @ 2M x 256 rows (atomic-bound,
// rocprof 2026-09-12); REPLICAS independent [k, d] buffers (one per block
// group, ~16 MB total) cut per-cell contention by the replica count, then a
// trivial f32x4 pass folds replicas (guide G12: pre-aggregate, then atomics).

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

__global__ void reduce_rows_by_key_kernel(const float* __restrict__ x,
                                          const int* __restrict__ keys,
                                          float* __restrict__ work,
                                          long long n_rows, long long d,
                                          long long kd, int replicas) {
  const long long waves_per_block = blockDim.x / RAFT_AMD_WAVE;
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  float* my = work + (long long)(blockIdx.x % replicas) * kd;
  long long row = (long long)blockIdx.x * waves_per_block + threadIdx.x / RAFT_AMD_WAVE;
  const long long stride = (long long)gridDim.x * waves_per_block;
  for (; row < n_rows; row += stride) {
    const int key = keys[row];
    const float* rp = x + row * d;
    float* sp = my + (long long)key * d;
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

// sort-based segmented accumulation: rows pre-ordered by key (perm from a
// radix argsort). Each wave walks a contiguous chunk of the sorted order,
// accumulating rows of the SAME key in registers (4 f32/lane for d<=256-ish,
// strided for larger d) and issuing atomics only at run boundaries —
// ~k + n/chunk atomic bursts instead of n*d single-element atomics
// (measured: the naive kernel is atomic-issue-rate bound at ~77 G/s,
// 33 ms @ 10M x 256; this design needs ~2 passes of HBM instead).
template <int MAX_DREG>  // registers per lane for the accumulator
__global__ void reduce_rows_by_key_sorted_kernel(const float* __restrict__ x,
                                                 const int* __restrict__ perm,
                                                 const int* __restrict__ keys_sorted,
                                                 float* __restrict__ sums,
                                                 float* __restrict__ counts,
                                                 const float* __restrict__ dmin,
                                                 float* __restrict__ inertia_acc,
                                                 long long n_rows, long long d,
                                                 long long chunk) {
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const long long wave_id = ((long long)blockIdx.x * blockDim.x + threadIdx.x) / RAFT_AMD_WAVE;
  const long long start = wave_id * chunk;
  if (start >= n_rows) return;
  const long long stop = min(start + chunk, n_rows);

  const int dreg = (int)((d + RAFT_AMD_WAVE - 1) / RAFT_AMD_WAVE);
  float acc[MAX_DREG];
#pragma unroll
  for (int j = 0; j < MAX_DREG; j++) acc[j] = 0.f;

  int cur_key = keys_sorted[start];
  int run_len = 0;
  double local_inertia = 0.0;
  for (long long i = start; i < stop; i++) {
    const int key = keys_sorted[i];
    if (inertia_acc && lane == 0) local_inertia += (double)dmin[perm[i]];
    if (key != cur_key) {
      float* sp = sums + (long long)cur_key * d;
      for (int j = 0; j < dreg; j++) {
        const long long col = (long long)j * RAFT_AMD_WAVE + lane;
        if (col < d) atomicAdd(&sp[col], acc[j]);
        acc[j] = 0.f;
      }
      if (counts && lane == 0) atomicAdd(&counts[cur_key], (float)run_len);
      run_len = 0;
      cur_key = key;
    }
    const float* rp = x + (long long)perm[i] * d;
    for (int j = 0; j < dreg; j++) {
      const long long col = (long long)j * RAFT_AMD_WAVE + lane;
      if (col < d) acc[j] += rp[col];
    }
    run_len++;
  }
  float* sp = sums + (long long)cur_key * d;
  for (int j = 0; j < dreg; j++) {
    const long long col = (long long)j * RAFT_AMD_WAVE + lane;
    if (col < d) atomicAdd(&sp[col], acc[j]);
  }
  if (counts && lane == 0) atomicAdd(&counts[cur_key], (float)run_len);
  if (inertia_acc && lane == 0) atomicAdd(inertia_acc, (float)local_inertia);
}

// Fused centroid-update + exact-fp32 verify/refine in ONE X pass.
// Rows arrive key-sorted, so the chosen centroid row is run-constant and
// lives in registers; each row's x fragment is loaded once and used for BOTH
// the exact distance refinement and the centroid-sum accumulation. Rows
// inside the split-error margin rescan all centroids exactly (rare); a
// rescan that CHANGES the assignment routes that row's contribution to the
// new cluster via direct atomics (and fixes amin/dmin).
// VEC4: lane-contiguous column map col(j) = (j/4)*256 + lane*4 + j%4 so each
// lane's 4-column chunk loads as ONE dwordx4 (wave covers 1024 B/chunk);
// requires d % 4 == 0. Otherwise the classic col(j) = j*64 + lane scalar map.
template <int MAX_DREG, bool VEC4>
__global__ void kmeans_update_verify_kernel(
    const float* __restrict__ x, const int* __restrict__ perm,
    const int* __restrict__ keys_sorted, const float* __restrict__ c,
    const float* __restrict__ xn, float* __restrict__ dmin,
    int* __restrict__ amin, const float* __restrict__ dmin2,
    const float* __restrict__ cn_max_p, float* __restrict__ sums,
    float* __restrict__ counts, float* __restrict__ inertia_acc,
    long long n_rows, long long d, int n_centroids, long long chunk,
    float lead, float tail) {
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const long long wave_id =
      ((long long)blockIdx.x * blockDim.x + threadIdx.x) / RAFT_AMD_WAVE;
  const long long start = wave_id * chunk;
  if (start >= n_rows) return;
  const long long stop = min(start + chunk, n_rows);
  const float cn_max = *cn_max_p;

  float acc[MAX_DREG], creg[MAX_DREG], xv[MAX_DREG];
#pragma unroll
  for (int j = 0; j < MAX_DREG; j++) acc[j] = 0.f;
  int cur_key = -1;
  int run_len = 0;
  double local_inertia = 0.0;

  auto colmap = [&](int j) -> long long {
    if constexpr (VEC4)
      return (long long)(j >> 2) * (RAFT_AMD_WAVE * 4) + lane * 4 + (j & 3);
    else
      return (long long)j * RAFT_AMD_WAVE + lane;
  };
  auto load_row = [&](const float* __restrict__ rp, float* dst) {
    if constexpr (VEC4) {
      #pragma unroll
      for (int q = 0; q < (MAX_DREG + 3) / 4; q++) {
        const long long base = (long long)q * (RAFT_AMD_WAVE * 4) + lane * 4;
        if (base + 3 < d) {
          const float4 t = *reinterpret_cast<const float4*>(rp + base);
          dst[q * 4 + 0] = t.x; dst[q * 4 + 1] = t.y;
          dst[q * 4 + 2] = t.z; dst[q * 4 + 3] = t.w;
        } else {
          for (int jj = 0; jj < 4 && q * 4 + jj < MAX_DREG; jj++) {
            const long long col = base + jj;
            dst[q * 4 + jj] = col < d ? rp[col] : 0.f;
          }
        }
      }
    } else {
      #pragma unroll
      for (int j = 0; j < MAX_DREG; j++) {
        const long long col = colmap(j);
        dst[j] = col < d ? rp[col] : 0.f;
      }
    }
  };

  auto flush = [&]() {
    if (cur_key < 0) return;
    float* sp = sums + (long long)cur_key * d;
    #pragma unroll
    for (int j = 0; j < MAX_DREG; j++) {
      const long long col = colmap(j);
      if (col < d) atomicAdd(&sp[col], acc[j]);
      acc[j] = 0.f;
    }
    if (lane == 0 && run_len) atomicAdd(&counts[cur_key], (float)run_len);
    run_len = 0;
  };

  for (long long i = start; i < stop; i++) {
    const long long row = perm[i];
    const int key = keys_sorted[i];
    if (key != cur_key) {
      flush();
      cur_key = key;
      load_row(c + (long long)key * d, creg);
    }
    const float* rp = x + row * d;
    load_row(rp, xv);
    const float xnr = xn[row];
    const float margin = dmin2[row] - dmin[row];
    const float bound = 2.f * (lead * sqrtf(fmaxf(xnr * cn_max, 0.f)) +
                               tail * (xnr + cn_max));
    if (margin < bound) {
      // exact rescan over all centroids (rare: near-ties only)
      float bestv = INFINITY;
      int besti = 0;
      for (int jc = 0; jc < n_centroids; jc++) {
        const float* cp = c + (long long)jc * d;
        float a = 0.f;
        #pragma unroll
    for (int j = 0; j < MAX_DREG; j++) {
          const long long col = colmap(j);
          if (col < d) {
            const float diff = xv[j] - cp[col];
            a += diff * diff;
          }
        }
        a = wave_reduce_sum(a);
        if (a < bestv) { bestv = a; besti = jc; }
      }
      if (lane == 0) {
        dmin[row] = bestv;
        amin[row] = besti;
      }
      local_inertia += (double)bestv;
      if (besti == cur_key) {
        #pragma unroll
    for (int j = 0; j < MAX_DREG; j++) acc[j] += xv[j];
        run_len++;
      } else {
        // reassigned: direct atomics into the new cluster
        float* sp = sums + (long long)besti * d;
        #pragma unroll
    for (int j = 0; j < MAX_DREG; j++) {
          const long long col = colmap(j);
          if (col < d) atomicAdd(&sp[col], xv[j]);
        }
        if (lane == 0) atomicAdd(&counts[besti], 1.f);
      }
    } else {
      // exact fp32 refinement of the chosen distance (creg is run-resident)
      float a = 0.f;
      #pragma unroll
    for (int j = 0; j < MAX_DREG; j++) {
        const float diff = xv[j] - creg[j];
        a += diff * diff;
      }
      a = wave_reduce_sum(a);
      if (lane == 0) dmin[row] = a;
      local_inertia += (double)a;
      #pragma unroll
    for (int j = 0; j < MAX_DREG; j++) acc[j] += xv[j];
      run_len++;
    }
  }
  flush();
  if (inertia_acc && lane == 0) atomicAdd(inertia_acc, (float)local_inertia);
}

void launch_kmeans_update_verify(const float* x, const int* perm,
                                 const int* keys_sorted, const float* c,
                                 const float* xn, float* dmin, int* amin,
                                 const float* dmin2, const float* cn_max_dev,
                                 float* sums, float* counts, float* inertia_acc,
                                 long long n_rows,
                                 long long d, int n_centroids, hipStream_t stream,
                                 float lead, float tail) {
  const long long n_waves_target = 2048 * 4;
  long long chunk = (n_rows + n_waves_target - 1) / n_waves_target;
  if (chunk < 8) chunk = 8;
  const long long n_waves = (n_rows + chunk - 1) / chunk;
  const int grid = (int)((n_waves * RAFT_AMD_WAVE + 255) / 256);
#define KMUV_LAUNCH(MD, V4)                                                   \
  hipLaunchKernelGGL((kmeans_update_verify_kernel<MD, V4>), dim3(grid),       \
                     dim3(256), 0, stream, x, perm, keys_sorted, c, xn, dmin, \
                     amin, dmin2, cn_max_dev, sums, counts, inertia_acc,      \
                     n_rows, d, n_centroids, chunk, lead, tail)
  // VEC4 keeps only d/4 lanes loading (4 cols per lane): require d >= 256
  // so all 64 lanes stay active
  const bool v4 = (d % 4 == 0) && d >= 256;
  if (d <= 256) {
    if (v4) KMUV_LAUNCH(4, true); else KMUV_LAUNCH(4, false);
  } else {
    if (v4) KMUV_LAUNCH(16, true); else KMUV_LAUNCH(16, false);
  }
#undef KMUV_LAUNCH
}

// fused centroid prep: split fp32 centroids into bf16 slices + row sq-norms
// in ONE pass (replaces ~6 torch ops per k-means iteration).
__global__ void split_bf16_norms_kernel(const float* __restrict__ c,
                                        __bf16* __restrict__ s0,
                                        __bf16* __restrict__ s1,
                                        __bf16* __restrict__ s2,
                                        float* __restrict__ cn,
                                        float* __restrict__ cn_max,
                                        int nslice,
                                        long long n_rows, long long d) {
  __shared__ float lds[256 / RAFT_AMD_WAVE];
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const float* rp = c + row * d;
    float acc = 0.f;
    for (long long j = threadIdx.x; j < d; j += blockDim.x) {
      const float v = rp[j];
      acc += v * v;
      const __bf16 h0 = (__bf16)v;
      s0[row * d + j] = h0;
      if (nslice > 1) {
        const float r1 = v - (float)h0;
        const __bf16 h1 = (__bf16)r1;
        s1[row * d + j] = h1;
        if (nslice > 2) s2[row * d + j] = (__bf16)(r1 - (float)h1);
      }
    }
    acc = wave_reduce_sum(acc);
    const int wid = threadIdx.x / RAFT_AMD_WAVE, lane = threadIdx.x % RAFT_AMD_WAVE;
    if (lane == 0) lds[wid] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      float s = 0.f;
      for (int w = 0; w < (int)(blockDim.x / RAFT_AMD_WAVE); w++) s += lds[w];
      cn[row] = s;
      // fused max(cn) (norms are non-negative: int-ordered atomicMax works)
      if (cn_max) atomicMax(reinterpret_cast<int*>(cn_max), __float_as_int(s));
    }
    __syncthreads();
  }
}

void launch_split_bf16_norms(const float* c, void* s0, void* s1, void* s2, float* cn,
                             int nslice, long long n_rows, long long d,
                             hipStream_t stream, float* cn_max) {
  if (cn_max) (void)hipMemsetAsync(cn_max, 0, sizeof(float), stream);
  int grid = (int)(n_rows < 2048 ? n_rows : 2048);
  hipLaunchKernelGGL(split_bf16_norms_kernel, dim3(grid), dim3(256), 0, stream, c,
                     (__bf16*)s0, (__bf16*)s1, (__bf16*)s2, cn, cn_max, nslice,
                     n_rows, d);
}

// centroid update: c[k] = counts[k] > 0 ? sums[k]/counts[k] : c[k]
// (replaces the where/div/clamp torch op chain per iteration)
__global__ void kmeans_update_centroids_kernel(const float* __restrict__ sums,
                                               const float* __restrict__ counts,
                                               float* __restrict__ c, long long k,
                                               long long d) {
  const long long total = k * d;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const float cnt = counts[t / d];
    if (cnt > 0.f) c[t] = sums[t] / cnt;
  }
}

void launch_kmeans_update_centroids(const float* sums, const float* counts, float* c,
                                    long long k, long long d, hipStream_t stream) {
  hipLaunchKernelGGL(kmeans_update_centroids_kernel, dim3(grid_1d(k * d, 256)),
                     dim3(256), 0, stream, sums, counts, c, k, d);
}

// Exact-fp32 verification/repair pass for the split-bf16 fused L2-NN
// (the "verified bf16x2" engine): for every row, recompute the distance to
// the chosen centroid in exact fp32; rows whose (best, second-best) margin
// is inside the provable split-emulation error bound rescan ALL centroids
// in exact fp32 and repair the argmin. Device-side only — no host sync.
//
// Error bound, parametrized by (lead, tail) per split mode; the dot-product
// emulation error satisfies |Δdot| <= lead*sqrt(xn*cn) + tail*(xn+cn):
//   bf16x2v (2 slices, 3 products; |x - x0 - x1| <= 2^-18|x|):
//     dropped p11 + rounding <= ~2^-14.5*sqrt(xn*cn)
//     -> lead = 2^-13, tail = 2^-18 (~3x headroom + fp32 accumulation)
//   bf16x1v (1 slice, 1 product; |x - bf16(x)| <= 2^-9|x| on both sides):
//     |x.c - a.b| <= |δa.c| + |a.δb| <= 2.01*2^-9*||x||*||c||  (Cauchy-Schwarz)
//     plus MFMA fp32 accumulation (~d*2^-24, <= 2^-16 at d=256)
//     -> lead = 2^-7, tail = 2^-12 (~2.3x headroom)
// The kernel bound is 2*E because best may be over- and runner-up
// under-estimated by E each; margin >= 2E proves the emulated argmin exact.
__global__ void l2nn_verify_repair_kernel(const float* __restrict__ x,
                                          const float* __restrict__ c,
                                          const float* __restrict__ xn,
                                          float* __restrict__ dmin,
                                          int* __restrict__ amin,
                                          const float* __restrict__ dmin2,
                                          const float* __restrict__ cn_max_p,
                                          long long m, int n, int d,
                                          float lead, float tail) {
  const float cn_max = *cn_max_p;
  const long long waves_per_block = blockDim.x / RAFT_AMD_WAVE;
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  long long row = (long long)blockIdx.x * waves_per_block + threadIdx.x / RAFT_AMD_WAVE;
  const long long stride = (long long)gridDim.x * waves_per_block;
  for (; row < m; row += stride) {
    const float* rp = x + row * d;
    const float xnr = xn[row];
    const float margin = dmin2[row] - dmin[row];
    const float bound = 2.f * (lead * sqrtf(fmaxf(xnr * cn_max, 0.f)) +
                               tail * (xnr + cn_max));
    int a = amin[row];
    if (margin < bound) {
      // full exact rescan (rare: near-ties only)
      float bestv = INFINITY;
      int besti = 0;
      for (int j = 0; j < n; j++) {
        const float* cp = c + (long long)j * d;
        float acc = 0.f;
        for (int t = lane; t < d; t += RAFT_AMD_WAVE) {
          const float diff = rp[t] - cp[t];
          acc += diff * diff;
        }
        acc = wave_reduce_sum(acc);
        if (acc < bestv) { bestv = acc; besti = j; }
      }
      if (lane == 0) {
        dmin[row] = bestv;
        amin[row] = besti;
      }
    } else {
      // exact fp32 refinement of the chosen distance (float4-vectorized:
      // one 16B load per lane covers d=256 in a single step)
      const float* cp = c + (long long)a * d;
      float acc = 0.f;
      const int d4 = d / 4;
      const float4* rp4 = reinterpret_cast<const float4*>(rp);
      const float4* cp4 = reinterpret_cast<const float4*>(cp);
      for (int t = lane; t < d4; t += RAFT_AMD_WAVE) {
        const float4 xv = rp4[t];
        const float4 cv = cp4[t];
        const float d0 = xv.x - cv.x, d1 = xv.y - cv.y;
        const float d2_ = xv.z - cv.z, d3 = xv.w - cv.w;
        acc += d0 * d0 + d1 * d1 + d2_ * d2_ + d3 * d3;
      }
      for (int t = d4 * 4 + lane; t < d; t += RAFT_AMD_WAVE) {
        const float diff = rp[t] - cp[t];
        acc += diff * diff;
      }
      acc = wave_reduce_sum(acc);
      if (lane == 0) dmin[row] = acc;
    }
  }
}

void launch_l2nn_verify_repair(const float* x, const float* c, const float* xn,
                               float* dmin, int* amin, const float* dmin2,
                               const float* cn_max_dev, long long m, int n, int d,
                               hipStream_t stream, float lead, float tail) {
  // uncapped: one wave per row (row-serial grid-stride was latency-bound)
  long long blocks = (m * RAFT_AMD_WAVE + 255) / 256;
  int grid = (int)(blocks > 2147483647ll ? 2147483647ll : blocks);
  hipLaunchKernelGGL(l2nn_verify_repair_kernel, dim3(grid), dim3(256), 0, stream,
                     x, c, xn, dmin, amin, dmin2, cn_max_dev, m, n, d, lead, tail);
}

void launch_reduce_rows_by_key_sorted(const float* x, const int* perm,
                                      const int* keys_sorted, float* sums,
                                      float* counts, const float* dmin,
                                      float* inertia_acc, long long n_rows,
                                      long long d, hipStream_t stream) {
  // chunk sized so the grid fills the chip (~2048 blocks * 4 waves)
  const long long n_waves_target = 2048 * 4;
  long long chunk = (n_rows + n_waves_target - 1) / n_waves_target;
  if (chunk < 8) chunk = 8;
  const long long n_waves = (n_rows + chunk - 1) / chunk;
  const int grid = (int)((n_waves * RAFT_AMD_WAVE + 255) / 256);
  if (d <= 256) {
    hipLaunchKernelGGL((reduce_rows_by_key_sorted_kernel<4>), dim3(grid), dim3(256),
                       0, stream, x, perm, keys_sorted, sums, counts, dmin,
                       inertia_acc, n_rows, d, chunk);
  } else if (d <= 1024) {
    hipLaunchKernelGGL((reduce_rows_by_key_sorted_kernel<16>), dim3(grid), dim3(256),
                       0, stream, x, perm, keys_sorted, sums, counts, dmin,
                       inertia_acc, n_rows, d, chunk);
  } else {
    hipLaunchKernelGGL((reduce_rows_by_key_sorted_kernel<64>), dim3(grid), dim3(256),
                       0, stream, x, perm, keys_sorted, sums, counts, dmin,
                       inertia_acc, n_rows, d, chunk);
  }
}

__global__ void fold_replicas_kernel(const float* __restrict__ work,
                                     float* __restrict__ out, long long kd,
                                     int replicas) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < kd;
       i += stride) {
    float acc = 0.f;
    for (int r = 0; r < replicas; r++) acc += work[(long long)r * kd + i];
    out[i] = acc;
  }
}

void launch_reduce_rows_by_key(const float* x, const int* keys, float* work,
                               float* out, long long n_rows, long long d,
                               long long n_keys, int replicas, hipStream_t stream) {
  const long long kd = n_keys * d;
  int grid = grid_1d(n_rows * RAFT_AMD_WAVE, 256);
  hipLaunchKernelGGL(reduce_rows_by_key_kernel, dim3(grid), dim3(256), 0, stream,
                     x, keys, work, n_rows, d, kd, replicas);
  hipLaunchKernelGGL(fold_replicas_kernel, dim3(grid_1d(kd, 256)), dim3(256), 0,
                     stream, work, out, kd, replicas);
}

}  // namespace raft_amd
