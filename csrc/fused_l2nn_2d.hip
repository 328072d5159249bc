// Fused L2-NN, 2D-grid variant: one block per (row-tile, col-tile-GROUP) with
// an XCD-contiguous block swizzle + a per-row partials-combine kernel.
//
// WHY (measured, profiles/pmc_fused_l2nn_tcc_l2.txt): the v1 kernel loops all
// col-tiles inside one block, so every block re-stages its 131 KB X tile once
// per col-tile; the per-XCD working set (64 resident blocks x 131 KB) blows
// the 4 MiB per-XCD L2 -> 48.6% hit rate, HBM reads == full 8x X re-read.
// Here the col-tiles of a row tile are TEMPORALLY ADJACENT ON ONE XCD
// (guide T1: default dispatch round-robins blockIdx across the 8 XCDs, so we
// remap bijectively so XCD x executes a contiguous range of row-major tile
// indices): the X tile is read into that XCD's L2 once and the other
// col-tile groups hit (measured 93.2% hit rate, HBM 49 GB -> 6.7 GB,
// profiles/pmc_fused_l2nn_2d_tcc.txt). Each block emits per-row
// (best, second, argmin) PARTIALS for its group; a bandwidth-bound combine
// kernel merges the n/128/GT partials per row (disjoint column ranges -> a
// plain top-2 merge) and applies ||x||^2.
//
// GT (col-tiles per block) trades L2 working set against epilogue
// amortization: GT=1 minimizes the per-XCD X footprint (64 resident blocks
// share 8 row tiles) but pays the reduce epilogue per tile; GT=4 pays it
// once per 4 tiles but doubles the X footprint. The epilogue itself is an
// LDS-transpose serial merge (each of 128 threads merges its row's 32 lane
// candidates) instead of the 12-shuffle butterfly — cheaper, and it
// subsumes the column-half-wave combine.
//
// Reference parity: same fused-L2-NN contract as fused_l2nn.hip (RAFT's
// fusedL2NN / k-means assignment step).

#include <hip/hip_runtime.h>

#include "mfma_common.h"

namespace raft_amd {

bool l2nn_phased();  // fused_l2nn.hip: RAFT_AMD_L2NN_PHASED

template <int NSLICE, int GT, bool PHASED = false, bool ADIRECT = false>
__launch_bounds__(256, 2)
__global__ void fused_l2nn_2d_kernel(const __bf16* __restrict__ x0,
                                     const __bf16* __restrict__ x1,
                                     const __bf16* __restrict__ x2,
                                     const __bf16* __restrict__ c0,
                                     const __bf16* __restrict__ c1,
                                     const __bf16* __restrict__ c2,
                                     const float* __restrict__ cn,
                                     float* __restrict__ pd,
                                     float* __restrict__ pd2,
                                     int* __restrict__ pi,
                                     long long m, int n, int d, int n_groups) {
  extern __shared__ __bf16 smem[];
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    if constexpr (ADIRECT) {
      cs[s] = smem + s * 8192;   // C-only LDS; X reads are direct
      xs[s] = smem;
    } else {
      xs[s] = smem + s * 8192;
      cs[s] = smem + (NSLICE + s) * 8192;
    }
  }

  // bijective XCD-contiguous remap (guide T1 / m204 variant): XCD x = bid%8
  // executes slots bid/8 over a contiguous band of row-major tile indices,
  // so its resident blocks share X row-tiles through the per-XCD L2.
  const int nwg = gridDim.x;
  const int bid = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, slot = bid >> 3;
  const int t = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  const long long row0 = (long long)(t / n_groups) * 128;
  const int grp = t % n_groups;

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;  // 2x2 wave grid

  float best[4][4], best2[4][4];
  int bidx[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) {
      best[a][b] = INFINITY;
      best2[a][b] = INFINITY;
      bidx[a][b] = 0;
    }

  // NOT unrolled: the k-loop body is large; unrolling it GT times (8 at the
  // 1-slice default) bloats I-cache for zero register benefit
#pragma unroll 1
  for (int g = 0; g < GT; g++) {
    const long long col0 = ((long long)grp * GT + g) * 128;
    f32x4 acc[4][4];
#pragma unroll
    for (int a = 0; a < 4; a++)
#pragma unroll
      for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    if constexpr (ADIRECT) {
      mfma_tile_kloop_ad<NSLICE>(xg, cg, cs, acc, row0, col0, d, m - 1, n - 1,
                                 wr, wc, lane);
    } else if constexpr (PHASED && NSLICE == 2) {
      mfma_tile_kloop_p2(xg, cg, xs, cs, acc, row0, col0, d, m - 1, n - 1, wr,
                         wc, lane);
    } else {
      mfma_tile_kloop<NSLICE>(xg, cg, xs, cs, acc, row0, col0, d, m - 1, n - 1,
                              wr, wc, lane);
    }

    // lane-local running top-2 across the group's tiles (disjoint columns);
    // cn[col] depends only on fc -> 4 loads per tile instead of 64
    const int col_base = (int)col0 + wc * 64;
    float cn_r[4];
#pragma unroll
    for (int fc = 0; fc < 4; fc++) cn_r[fc] = cn[col_base + fc * 16 + (lane & 15)];
#pragma unroll
    for (int fr = 0; fr < 4; fr++) {
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int col = col_base + fc * 16 + (lane & 15);
          const float s = cn_r[fc] - 2.f * acc[fr][fc][reg];
          // new second-best = middle of {s, best, best2} (invariant
          // best <= best2): one v_med3_f32 instead of a cndmask chain
          best2[fr][reg] = __builtin_amdgcn_fmed3f(s, best[fr][reg],
                                                   best2[fr][reg]);
          if (s < best[fr][reg]) {
            best[fr][reg] = s;
            bidx[fr][reg] = col;
          }
        }
      }
    }
  }

  // LDS-transpose epilogue: each thread parks its 16 (fr,reg) top-2 triples;
  // then one thread per row serially merges the row's 32 lane-candidates
  // (both column-half waves at once — subsumes the wc-combine). Stride 33
  // keeps the 32-candidate rows off a single bank.
  __syncthreads();
  float* lv = reinterpret_cast<float*>(smem);   // [128][33]
  float* lv2 = lv + 128 * 33;                   // [128][33]
  int* li = reinterpret_cast<int*>(lv2 + 128 * 33);
  const int cand = wc * 16 + (lane & 15);
#pragma unroll
  for (int fr = 0; fr < 4; fr++)
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      const int rl = wr * 64 + fr * 16 + ((lane >> 4) & 3) * 4 + reg;
      lv[rl * 33 + cand] = best[fr][reg];
      lv2[rl * 33 + cand] = best2[fr][reg];
      li[rl * 33 + cand] = bidx[fr][reg];
    }
  __syncthreads();
  const int rl = threadIdx.x;
  if (rl < 128) {
    const long long row = row0 + rl;
    if (row < m) {
      float v = INFINITY, v2 = INFINITY;
      int vi = 0;
#pragma unroll 8
      for (int c = 0; c < 32; c++) {
        const float b = lv[rl * 33 + c];
        const float b2 = lv2[rl * 33 + c];
        const int bi = li[rl * 33 + c];
        v2 = fminf(fminf(v2, b2), fmaxf(v, b));  // columns all distinct
        if (b < v || (b == v && bi < vi)) { v = b; vi = bi; }
      }
      const long long o = (long long)grp * m + row;
      pd[o] = v;
      pd2[o] = v2;
      pi[o] = vi;
    }
  }
}

// BK=32 / 4-blocks-per-CU variant: 8 KiB LDS tiles (32 KiB total for
// 2 slices) double the resident blocks per CU, so 4 independent load-drains
// interleave per SIMD instead of 2. Epilogue is the v1-style shuffle reduce
// (once per GT tiles) + tiny LDS half-combine — the [128][33] transpose
// epilogue would need 50 KiB and cap occupancy back at 2.
template <int NSLICE, int GT>
__launch_bounds__(256, 4)
__global__ void fused_l2nn_2d_bk32_kernel(const __bf16* __restrict__ x0,
                                          const __bf16* __restrict__ x1,
                                          const __bf16* __restrict__ x2,
                                          const __bf16* __restrict__ c0,
                                          const __bf16* __restrict__ c1,
                                          const __bf16* __restrict__ c2,
                                          const float* __restrict__ cn,
                                          float* __restrict__ pd,
                                          float* __restrict__ pd2,
                                          int* __restrict__ pi,
                                          long long m, int n, int d,
                                          int n_groups) {
  extern __shared__ __bf16 smem[];
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 4096;
    cs[s] = smem + (NSLICE + s) * 4096;
  }
  const int nwg = gridDim.x;
  const int bid = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, slot = bid >> 3;
  const int t = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  const long long row0 = (long long)(t / n_groups) * 128;
  const int grp = t % n_groups;

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;

  float best[4][4], best2[4][4];
  int bidx[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) {
      best[a][b] = INFINITY;
      best2[a][b] = INFINITY;
      bidx[a][b] = 0;
    }

  // NOT unrolled: the k-loop body is large; unrolling it GT times (8 at the
  // 1-slice default) bloats I-cache for zero register benefit
#pragma unroll 1
  for (int g = 0; g < GT; g++) {
    const long long col0 = ((long long)grp * GT + g) * 128;
    f32x4 acc[4][4];
#pragma unroll
    for (int a = 0; a < 4; a++)
#pragma unroll
      for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};
    mfma_tile_kloop_s32<NSLICE>(xg, cg, xs, cs, acc, row0, col0, d, m - 1,
                                n - 1, wr, wc, lane);
    const int col_base = (int)col0 + wc * 64;
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++)
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int col = col_base + fc * 16 + (lane & 15);
          const float s = cn[col] - 2.f * acc[fr][fc][reg];
          // new second-best = middle of {s, best, best2} (invariant
          // best <= best2): one v_med3_f32 instead of a cndmask chain
          best2[fr][reg] = __builtin_amdgcn_fmed3f(s, best[fr][reg],
                                                   best2[fr][reg]);
          if (s < best[fr][reg]) {
            best[fr][reg] = s;
            bidx[fr][reg] = col;
          }
        }
  }

  // ONE cross-lane top-2 reduce (index-aware), then LDS half-combine
#pragma unroll
  for (int fr = 0; fr < 4; fr++)
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      float v = best[fr][reg], v2 = best2[fr][reg];
      int vi = bidx[fr][reg];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) {
        const float ov = __shfl_xor(v, off, RAFT_AMD_WAVE);
        const float ov2 = __shfl_xor(v2, off, RAFT_AMD_WAVE);
        const int oi = __shfl_xor(vi, off, RAFT_AMD_WAVE);
        float new2 = fminf(v2, ov2);
        if (oi != vi) new2 = fminf(new2, fmaxf(v, ov));
        v2 = new2;
        if (ov < v || (ov == v && oi < vi)) { v = ov; vi = oi; }
      }
      best[fr][reg] = v;
      best2[fr][reg] = v2;
      bidx[fr][reg] = vi;
    }
  __syncthreads();
  float* comb_v = reinterpret_cast<float*>(smem);
  int* comb_i = reinterpret_cast<int*>(comb_v + 256);
  float* comb_v2 = reinterpret_cast<float*>(comb_i + 256);
  if ((lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;
        comb_v[wc * 128 + rl] = best[fr][reg];
        comb_i[wc * 128 + rl] = bidx[fr][reg];
        comb_v2[wc * 128 + rl] = best2[fr][reg];
      }
  }
  __syncthreads();
  const int rl = threadIdx.x;
  if (rl < 128) {
    const long long row = row0 + rl;
    if (row < m) {
      const float v0 = comb_v[rl], v1 = comb_v[128 + rl];
      const int i0 = comb_i[rl], i1 = comb_i[128 + rl];
      const float s2 = fminf(fminf(comb_v2[rl], comb_v2[128 + rl]),
                             fmaxf(v0, v1));
      const bool take1 = (v1 < v0) || (v1 == v0 && i1 < i0);
      const long long o = (long long)grp * m + row;
      pd[o] = take1 ? v1 : v0;
      pi[o] = take1 ? i1 : i0;
      pd2[o] = s2;
    }
  }
}

// Merge the n_groups partials per row; groups cover disjoint column ranges
// so the cross-group second-best is the plain top-2 merge of independent
// lists. Adds ||x||^2 once and clamps the best (matching v1's write).
__global__ void l2nn_combine_partials_kernel(const float* __restrict__ pd,
                                             const float* __restrict__ pd2,
                                             const int* __restrict__ pi,
                                             const float* __restrict__ xn,
                                             float* __restrict__ dmin,
                                             int* __restrict__ amin,
                                             float* __restrict__ dmin2,
                                             long long m, int n_groups) {
  const long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= m) return;
  float v = INFINITY, v2 = INFINITY;
  int vi = 0;
  for (int g = 0; g < n_groups; g++) {
    const long long o = (long long)g * m + row;
    const float b = pd[o], b2 = pd2[o];
    const int bi = pi[o];
    v2 = fminf(fminf(v2, b2), fmaxf(v, b));
    if (b < v || (b == v && bi < vi)) { v = b; vi = bi; }
  }
  const float x = xn[row];
  dmin[row] = fmaxf(v + x, 0.f);
  amin[row] = vi;
  if (dmin2) dmin2[row] = v2 + x;
}

// shared combine launch (also used by the 256^2 engine in fused_l2nn_256.hip)
void launch_l2nn_combine(const float* pd, const float* pd2, const int* pi,
                         const float* xn, float* dmin, int* amin, float* dmin2,
                         long long m, int n_groups, hipStream_t stream) {
  const long long cgrid = (m + 255) / 256;
  hipLaunchKernelGGL(l2nn_combine_partials_kernel, dim3((int)cgrid), dim3(256),
                     0, stream, pd, pd2, pi, xn, dmin, amin, dmin2, m,
                     n_groups);
}

static int l2nn_2d_gt_env() {
  static const int gt = [] {
    const char* e = getenv("RAFT_AMD_L2NN_GT");
    const int v = e ? atoi(e) : 0;
    return (v == 1 || v == 2 || v == 4 || v == 8) ? v : 0;
  }();
  return gt;
}

// effective col-tiles-per-block: nslice-aware default (measured at 10M x 256
// k=1024: 1-slice GT8 9.55 / GT4 9.69 / GT2 11.30 ms — the epilogue is a
// bigger share of the cheaper 1-product k-loop, so amortizing it over more
// tiles wins; 2-slice GT2 remains the optimum). Halve until it divides the
// col-tile
// count, so odd tile counts (e.g. n=384) keep the 2D engine's L2 win
// instead of falling back to v1 (round-1 NOTES item 15).
static int l2nn_2d_gt_for(int n, int nslice) {
  int gt = l2nn_2d_gt_env();
  if (gt == 0) gt = nslice == 1 ? 8 : 2;
  const int tiles = n / 128;
  while (gt > 1 && tiles % gt != 0) gt >>= 1;
  return gt;
}

bool fused_l2nn_2d_supported(int nslice, long long m, int n, int d) {
  static const char mode = [] {
    const char* e = getenv("RAFT_AMD_L2NN_2D");
    return e ? e[0] : 'a';
  }();
  if (mode == '0') return false;
  if (nslice > 2) return false;  // 3-slice LDS (96 KiB) drops to 1 block/CU
  if (n % 128 != 0) return false;
  if (mode == '1') return true;
  // auto: pays off when the col-tile loop is long enough that X re-reads
  // dominate and m is large enough that partial traffic amortizes
  // (measured: 16.9 ms vs v1's 17.4-18.0 at 10M x 256 k=1024, with HBM
  // reads 49 GB -> 6.7 GB; see BASELINE.md schedule table).
  return n >= 512 && m >= 1000000;
}

void launch_fused_l2nn_2d(const void** xsl, const void** csl, const float* xn,
                          const float* cn, float* pd, float* pd2, int* pi,
                          float* dmin, int* amin, float* dmin2,
                          long long m, int n, int d, int nslice,
                          hipStream_t stream) {
  static const bool bk32_env = [] {
    const char* e = getenv("RAFT_AMD_L2NN_BK32");
    return e && e[0] == '1';
  }();
  int gt = l2nn_2d_gt_for(n, nslice);
  // the BK32 variant is only instantiated up to GT=4: clamp BEFORE n_groups
  // so grid coverage matches the kernel's per-block tile count
  if (bk32_env && gt > 4) gt = 4;
  const int n_row_tiles = (int)((m + 127) / 128);
  const int n_groups = n / 128 / gt;
  const int grid = n_row_tiles * n_groups;
  static const bool adirect = [] {
    const char* e = getenv("RAFT_AMD_L2NN_AD");
    return e && e[0] == '1';
  }();
  // kloop needs NSLICE*2*16KiB (C-only when A-direct); the epilogue reuses
  // it ([128][33] f32 x2 + i32)
  const size_t lds_kloop = (size_t)nslice * (adirect ? 1 : 2) * 8192 * sizeof(__bf16);
  const size_t lds = lds_kloop > 3 * 128 * 33 * 4 ? lds_kloop : 3 * 128 * 33 * 4;
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
#define L2NN2D_LAUNCH(NS, G, P)                                               \
  do {                                                                         \
    if (adirect)                                                               \
      hipLaunchKernelGGL((fused_l2nn_2d_kernel<NS, G, false, true>),           \
                         dim3(grid), dim3(256), lds, stream, x0, x1, x2, c0,   \
                         c1, c2, cn, pd, pd2, pi, m, n, d, n_groups);          \
    else                                                                       \
      hipLaunchKernelGGL((fused_l2nn_2d_kernel<NS, G, P>), dim3(grid),         \
                         dim3(256), lds, stream, x0, x1, x2, c0, c1, c2, cn,   \
                         pd, pd2, pi, m, n, d, n_groups);                      \
  } while (0)
  if (bk32_env && (nslice == 1 || nslice == 2) && d % 32 == 0) {
    const size_t lds32 = (size_t)nslice * 2 * 4096 * sizeof(__bf16);
#define L2NN2D_BK32(NS, G)                                                     \
  hipLaunchKernelGGL((fused_l2nn_2d_bk32_kernel<NS, G>), dim3(grid),           \
                     dim3(256), lds32, stream, x0, x1, x2, c0, c1, c2, cn, pd, \
                     pd2, pi, m, n, d, n_groups)
    if (nslice == 1) {
      if (gt == 1) L2NN2D_BK32(1, 1);
      else if (gt == 2) L2NN2D_BK32(1, 2);
      else L2NN2D_BK32(1, 4);
    } else {
      if (gt == 1) L2NN2D_BK32(2, 1);
      else if (gt == 2) L2NN2D_BK32(2, 2);
      else L2NN2D_BK32(2, 4);
    }
#undef L2NN2D_BK32
    const long long cgrid32 = (m + 255) / 256;
    hipLaunchKernelGGL(l2nn_combine_partials_kernel, dim3((int)cgrid32),
                       dim3(256), 0, stream, pd, pd2, pi, xn, dmin, amin,
                       dmin2, m, n_groups);
    return;
  }
  const bool ph = l2nn_phased();
  if (nslice == 1) {
    if (gt == 1) L2NN2D_LAUNCH(1, 1, false);
    else if (gt == 2) L2NN2D_LAUNCH(1, 2, false);
    else if (gt == 4) L2NN2D_LAUNCH(1, 4, false);
    else L2NN2D_LAUNCH(1, 8, false);
  } else if (nslice == 2) {
    if (gt == 1) { if (ph) L2NN2D_LAUNCH(2, 1, true); else L2NN2D_LAUNCH(2, 1, false); }
    else if (gt == 2) { if (ph) L2NN2D_LAUNCH(2, 2, true); else L2NN2D_LAUNCH(2, 2, false); }
    else { if (ph) L2NN2D_LAUNCH(2, 4, true); else L2NN2D_LAUNCH(2, 4, false); }
  } else {
    throw std::runtime_error("fused_l2nn_2d: nslice must be 1 or 2");
  }
#undef L2NN2D_LAUNCH
  const long long cgrid = (m + 255) / 256;
  hipLaunchKernelGGL(l2nn_combine_partials_kernel, dim3((int)cgrid), dim3(256),
                     0, stream, pd, pd2, pi, xn, dmin, amin, dmin2, m,
                     n_groups);
}

}  // namespace raft_amd
