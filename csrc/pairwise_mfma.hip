// Pairwise L2-expanded distance tile kernel — split-bf16 MFMA with the
// epilogue fused into the C-write (BASELINE config 2's engine).
//
// Reference parity (WHAT): RAFT's historical pairwise-distance L2-expanded
// epilogue over the contraction engine. Versus the rocBLAS 3/6-GEMM
// beta-accumulate path, this writes the fp32 distance tile EXACTLY ONCE
// (d2 = xn + yn - 2 x.c computed in registers from the accumulator), cutting
// the dominant HBM cost of large distance matrices by ~5x
// (3-GEMM path: 3 writes + 2 reads + epilogue read/write = 7 tile passes).
//
// Same geometry/staging as fused_l2nn.hip (see mfma_common.h).

#include <hip/hip_runtime.h>

#include "mfma_common.h"

namespace raft_amd {

template <int NSLICE, bool BK32 = false>
__launch_bounds__(256, 2)
__global__ void pairwise_l2_kernel(const __bf16* __restrict__ x0,
                                   const __bf16* __restrict__ x1,
                                   const __bf16* __restrict__ x2,
                                   const __bf16* __restrict__ c0,
                                   const __bf16* __restrict__ c1,
                                   const __bf16* __restrict__ c2,
                                   const float* __restrict__ xn,
                                   const float* __restrict__ yn,
                                   float* __restrict__ out,
                                   long long m, long long n, int d,
                                   long long ldo, int sqrt_out, int rg) {
  extern __shared__ __bf16 smem[];
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 8192;
    cs[s] = smem + (NSLICE + s) * 8192;
  }

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;
  long long rt_, ct_;
  xcd_supertile_decode(rg, &rt_, &ct_);
  if (rt_ * 128 >= m || ct_ * 128 >= n) return;  // super-tile ragged edge
  const long long row0 = rt_ * 128;
  const long long col0 = ct_ * 128;

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  if constexpr (BK32 && NSLICE <= 2) {
    // counted-vmcnt BK=32 dbuf K-loop at 2 blocks/CU: no full drains AND
    // cross-block overlap (A/B RAFT_AMD_PW_BK32=1)
    Mfma256BK32 st;
    mfma256_bk32_setup<128>(st, row0, col0, d, m - 1, n - 1, wr, wc, lane);
    mfma256_bk32_kloop<NSLICE, 128>(x0, x1, c0, c1, smem, st, acc, d / 32);
  } else {
    mfma_tile_kloop<NSLICE>(xg, cg, xs, cs, acc, row0, col0, d, m - 1, n - 1,
                            wr, wc, lane);
  }

  // fused epilogue: d2 = max(xn[r] + yn[c] - 2 acc, 0).
  // C/D layout: col = lane&15 (+fc*16), row = (lane>>4)*4 + reg (+fr*16).
  if (row0 + 128 <= m && col0 + 128 <= n && (ldo & 3) == 0) {
    // interior tile: stage through padded LDS ([128][132] fp32, 67.5 KiB —
    // fits 2 blocks/CU) and dump as coalesced NONTEMPORAL dwordx4 rows.
    // The direct path issues 64 scalar 4 B stores/thread each with 64-bit
    // row*ldo address math — measured 12:1 VALU:MFMA, epilogue-dominated
    // (418 Gdist/s vs the ~1.6 Tdist/s write roofline).
    __syncthreads();  // K-loop LDS is dead; reuse
    float* tile = reinterpret_cast<float*>(smem);  // [128][132]
    float yn_r[4];
#pragma unroll
    for (int fc = 0; fc < 4; fc++)
      yn_r[fc] = yn[col0 + wc * 64 + fc * 16 + (lane & 15)];
#pragma unroll
    for (int fr = 0; fr < 4; fr++) {
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;
        const float xv = xn[row0 + rl];
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int cl = wc * 64 + fc * 16 + (lane & 15);
          float v = fmaxf(xv + yn_r[fc] - 2.f * acc[fr][fc][reg], 0.f);
          if (sqrt_out) v = sqrtf(v);
          tile[rl * 132 + cl] = v;
        }
      }
    }
    __syncthreads();
    // 256 threads x float4: each round writes 8 rows (32 els/row = 8 thr)
    const int tr = threadIdx.x >> 5;        // 0..7: row within round
    const int tc = (threadIdx.x & 31) * 4;  // 0..124: col (float4)
    const float* src = tile + tr * 132 + tc;
    float* dst = out + (row0 + tr) * ldo + col0 + tc;
    const long long dstep = 8 * ldo;
#pragma unroll
    for (int rnd = 0; rnd < 16; rnd++) {
      const f32x4 v4 = *reinterpret_cast<const f32x4*>(src + rnd * 8 * 132);
      __builtin_nontemporal_store(
          v4, reinterpret_cast<f32x4*>(dst + rnd * dstep));
    }
    return;
  }
#pragma unroll
  for (int fr = 0; fr < 4; fr++) {
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      const long long row = row0 + wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;
      if (row >= m) continue;
      const float xv = xn[row];
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const long long col = col0 + wc * 64 + fc * 16 + (lane & 15);
        if (col < n) {
          float v = fmaxf(xv + yn[col] - 2.f * acc[fr][fc][reg], 0.f);
          if (sqrt_out) v = sqrtf(v);
          out[row * ldo + col] = v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// filtered top-k emission: same MFMA engine, but instead of writing the tile,
// emit only candidates with d2 <= thr[row] into per-row bounded buffers
// (the sample->threshold->filter kNN scheme: the m x n distance tile never
// touches HBM; expected emissions ~ O(k) per row).
// ---------------------------------------------------------------------------
template <int NSLICE>
__launch_bounds__(256, 2)
__global__ void pairwise_l2_filter_kernel(const __bf16* __restrict__ x0,
                                          const __bf16* __restrict__ x1,
                                          const __bf16* __restrict__ x2,
                                          const __bf16* __restrict__ c0,
                                          const __bf16* __restrict__ c1,
                                          const __bf16* __restrict__ c2,
                                          const float* __restrict__ xn,
                                          const float* __restrict__ yn,
                                          const float* __restrict__ thr,
                                          float* __restrict__ out_d,
                                          int* __restrict__ out_i,
                                          int* __restrict__ cnt, int cap,
                                          long long col_offset, long long m,
                                          long long n, int d, int rg) {
  extern __shared__ __bf16 smem[];
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 8192;
    cs[s] = smem + (NSLICE + s) * 8192;
  }

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;
  long long rt_, ct_;
  xcd_supertile_decode(rg, &rt_, &ct_);
  if (rt_ * 128 >= m || ct_ * 128 >= n) return;  // super-tile ragged edge
  const long long row0 = rt_ * 128;
  const long long col0 = ct_ * 128;

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  mfma_tile_kloop<NSLICE>(xg, cg, xs, cs, acc, row0, col0, d, m - 1, n - 1,
                          wr, wc, lane);

  // Epilogue VALU diet (same recipe as the l2nn 2d kernel, measured there
  // via profiles/pmc_l2nn_x1v_gt8.txt): yn[col] and the col<n guard depend
  // only on fc — hoisted to 4 registers (out-of-range columns carry +inf,
  // so they can never pass the threshold); the fmax moves into the rare
  // admitted-candidate store, with thr clamped to >= 0 once per row so the
  // unclamped compare admits the same set.
  const long long colb = col0 + wc * 64 + (lane & 15);
  float yn_r[4];
#pragma unroll
  for (int fc = 0; fc < 4; fc++) {
    const long long col = colb + fc * 16;
    yn_r[fc] = col < n ? yn[col] : INFINITY;
  }
#pragma unroll
  for (int fr = 0; fr < 4; fr++) {
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      const long long row = row0 + wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;
      if (row >= m) continue;
      const float xv = xn[row];
      const float t = fmaxf(thr[row], 0.f);
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const float s = xv + yn_r[fc] - 2.f * acc[fr][fc][reg];
        if (s <= t) {
          const int pos = atomicAdd(&cnt[row], 1);
          if (pos < cap) {
            out_d[row * cap + pos] = fmaxf(s, 0.f);
            out_i[row * cap + pos] = (int)(colb + fc * 16 + col_offset);
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Persistent-X filter (bf16 index, d <= 128): the X row panel (full depth,
// <= 32 KiB) stages ONCE per block and stays LDS-resident while the block
// walks CT col-tiles, double-buffering ONLY the 16 KiB C chunks with counted
// vmcnt (flight depth 1 chunk, no full drains). Halves the gloads per
// K-step and removes the X re-stage entirely; 64 KiB LDS keeps 2 blocks/CU
// (the l2nn persistent-X experiment lost to occupancy at 1 block/CU — this
// configuration keeps both the residency AND the 2-block overlap).
// Block decode groups same-col-group blocks on one XCD so the C panels are
// the L2-shared working set.
// ---------------------------------------------------------------------------
template <int CT = 8>
__launch_bounds__(256, 2)
__global__ void pairwise_l2_filter_px_kernel(
    const __bf16* __restrict__ x0g, const __bf16* __restrict__ c0g,
    const float* __restrict__ xn, const float* __restrict__ yn,
    const float* __restrict__ thr, float* __restrict__ out_d,
    int* __restrict__ out_i, int* __restrict__ cnt, int cap,
    long long col_offset, long long m, long long n, int d, int n_row_tiles) {
  extern __shared__ __bf16 smem[];
  const int kts = d / 64;

  // bijective XCD-contiguous remap, COLUMN-group-major (blocks sharing a C
  // group land on one XCD)
  const int nwg = gridDim.x;
  const int bid = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, slot = bid >> 3;
  const int t = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  const long long row0 = (long long)(t % n_row_tiles) * 128;
  const int ctg = t / n_row_tiles;
  const long long col_base = (long long)ctg * CT * 128;
  if (row0 >= m || col_base >= n) return;
  const int nct = (int)((n - col_base + 127) / 128) < CT
                      ? (int)((n - col_base + 127) / 128)
                      : CT;

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;

  // stage the X panel once: kts chunks of [128][64] at kt*8192 elements
  for (int kt = 0; kt < kts; kt++)
    mfma_stage_tile128(x0g, smem + kt * 8192, row0, (long long)kt * 64, d,
                       m - 1);
  __bf16* cb0 = smem + kts * 8192;   // two 16 KiB C chunk buffers
  __bf16* cb1 = cb0 + 8192;
  const int total = nct * kts;
  auto stage_c = [&](int s, __bf16* buf) {
    const long long c0 = col_base + (long long)(s / kts) * 128;
    mfma_stage_tile128(c0g, buf, c0, (long long)(s % kts) * 64, d, n - 1);
  };
  stage_c(0, cb0);
  if (total > 1) stage_c(1, cb1);
  // X (4*kts gloads) + chunk 0 must land; chunk 1 stays in flight
  if (total > 1) {
    asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
  }

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int s = 0; s < total; s++) {
    const int kt = s % kts;
    __bf16* cbuf = (s & 1) ? cb1 : cb0;
    const __bf16* xchunk = smem + kt * 8192;
    // ALL fragments for this chunk first (a/b for both kf), then a barrier
    // confirming every wave's reads are done, THEN the refill of this
    // buffer with chunk s+2 — a mid-read refill races other waves' ds_reads
    bf16x8 a_frag[2][4], b_frag[2][4];
#pragma unroll
    for (int kf = 0; kf < 2; kf++) {
      const int kbyte = (kf * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
      for (int fr = 0; fr < 4; fr++) {
        const int rr = wr * 64 + fr * 16 + (lane & 15);
        a_frag[kf][fr] = *reinterpret_cast<const bf16x8*>(
            (const char*)xchunk + mfma_swz(rr * 128 + kbyte));
      }
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const int cc = wc * 64 + fc * 16 + (lane & 15);
        b_frag[kf][fc] = *reinterpret_cast<const bf16x8*>(
            (const char*)cbuf + mfma_swz(cc * 128 + kbyte));
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)\n\ts_barrier" ::: "memory");
    if (s + 2 < total) stage_c(s + 2, cbuf);
#pragma unroll
    for (int kf = 0; kf < 2; kf++)
#pragma unroll
      for (int fr = 0; fr < 4; fr++)
#pragma unroll
        for (int fc = 0; fc < 4; fc++)
          acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[kf][fr], b_frag[kf][fc], acc[fr][fc], 0, 0, 0);
    if (kt == kts - 1) {
      // filter epilogue for this col tile (registers + global atomics only)
      const long long col0 = col_base + (long long)(s / kts) * 128;
      // hoisted epilogue (see pairwise_l2_filter_kernel): yn + col guard
      // live in 4 regs; +inf padding makes out-of-range columns inadmissible
      const long long colb = col0 + wc * 64 + (lane & 15);
      float yn_r[4];
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const long long col = colb + fc * 16;
        yn_r[fc] = col < n ? yn[col] : INFINITY;
      }
#pragma unroll
      for (int fr = 0; fr < 4; fr++) {
#pragma unroll
        for (int reg = 0; reg < 4; reg++) {
          const long long row = row0 + wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;
          if (row >= m) continue;
          const float xv = xn[row];
          const float tv = fmaxf(thr[row], 0.f);
#pragma unroll
          for (int fc = 0; fc < 4; fc++) {
            const float sd = xv + yn_r[fc] - 2.f * acc[fr][fc][reg];
            if (sd <= tv) {
              const int pos = atomicAdd(&cnt[row], 1);
              if (pos < cap) {
                out_d[row * cap + pos] = fmaxf(sd, 0.f);
                out_i[row * cap + pos] = (int)(colb + fc * 16 + col_offset);
              }
            }
          }
        }
      }
#pragma unroll
      for (int a = 0; a < 4; a++)
#pragma unroll
        for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
    // chunk s+1 must have landed before the next iteration reads it (leave
    // the 4 loads issued for s+2 in flight)
    if (s + 2 < total) {
      asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
    }
  }
}

void launch_pairwise_l2_filter(const void** xsl, const void** csl, const float* xn,
                               const float* yn, const float* thr, float* out_d,
                               int* out_i, int* cnt, int cap, long long col_offset,
                               long long m, long long n, int d, int nslice,
                               hipStream_t stream) {
  // Persistent-X A/B (RAFT_AMD_KNN_PX=1): 7.5k vs 9.9k q/s at the 100M
  // headline — the 1-chunk flight depth can't hide HBM latency in the
  // streaming regime and the per-chunk lockstep loses to the independent
  // full-drain blocks. OFF by default; kept as a recorded experiment.
  static const bool px = [] {
    const char* e = getenv("RAFT_AMD_KNN_PX");
    return e && e[0] == '1';
  }();
  if (px && nslice == 1 && d % 64 == 0 && d <= 128 && n >= 1024) {
    constexpr int CT = 8;
    const int n_row_tiles = (int)((m + 127) / 128);
    const int n_groups = (int)((n + (long long)CT * 128 - 1) / ((long long)CT * 128));
    dim3 grid((unsigned)((long long)n_row_tiles * n_groups));
    const size_t lds = (size_t)(d / 64) * 8192 * 2 + 2 * 16384;  // X + 2 C bufs
    hipLaunchKernelGGL((pairwise_l2_filter_px_kernel<CT>), grid, dim3(256), lds,
                       stream, (const __bf16*)xsl[0], (const __bf16*)csl[0],
                       xn, yn, thr, out_d, out_i, cnt, cap, col_offset, m, n,
                       d, n_row_tiles);
    return;
  }
  const int rg = (int)((m + 1023) / 1024);       // ceil(R/8), R=ceil(m/128)
  const int cg = (int)((n + 1023) / 1024);
  dim3 grid((unsigned)((long long)rg * cg * 64));
  const size_t lds = (size_t)nslice * 2 * 8192 * sizeof(__bf16);
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  switch (nslice) {
    case 1:
      hipLaunchKernelGGL((pairwise_l2_filter_kernel<1>), grid, dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, yn, thr, out_d, out_i, cnt, cap,
                         col_offset, m, n, d, rg);
      break;
    case 2:
      hipLaunchKernelGGL((pairwise_l2_filter_kernel<2>), grid, dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, yn, thr, out_d, out_i, cnt, cap,
                         col_offset, m, n, d, rg);
      break;
    case 3: {
      static bool attr_set3 = false;
      if (!attr_set3) {
        HIP_CHECK(hipFuncSetAttribute((const void*)&pairwise_l2_filter_kernel<3>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      96 * 1024));
        attr_set3 = true;
      }
      hipLaunchKernelGGL((pairwise_l2_filter_kernel<3>), grid, dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, yn, thr, out_d, out_i, cnt, cap,
                         col_offset, m, n, d, rg);
      break;
    }
    default:
      throw std::runtime_error("pairwise_l2_filter: nslice must be 1, 2 or 3");
  }
}

void launch_pairwise_l2_mfma(const void** xsl, const void** csl, const float* xn,
                             const float* yn, float* out, long long m, long long n,
                             int d, long long ldo, int nslice, bool sqrt_out,
                             hipStream_t stream) {
  int rg = (int)((m + 1023) / 1024);             // ceil(R/8), R=ceil(m/128)
  const int cg = (int)((n + 1023) / 1024);
  dim3 grid((unsigned)((long long)rg * cg * 64));
  // column-major supertile slot order (adjacent output segments written by
  // temporally adjacent blocks — DRAM page locality); rg<0 flags it to
  // xcd_supertile_decode. RAFT_AMD_PW_RM=1 restores row-major for A/B.
  static const bool rowmajor = [] {
    const char* e = getenv("RAFT_AMD_PW_RM");
    return e && e[0] == '1';
  }();
  if (!rowmajor) rg = -rg;
  static const bool bk32 = [] {
    const char* e = getenv("RAFT_AMD_PW_BK32");
    return e && e[0] == '1';
  }();
  const size_t lds_k = (size_t)nslice * 2 * 8192 * sizeof(__bf16);
  const size_t lds_epi = 128 * 132 * 4;  // padded fp32 staging tile
  const size_t lds = lds_k > lds_epi ? lds_k : lds_epi;
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  switch (nslice) {
    case 1:
      if (bk32 && d % 32 == 0)
        hipLaunchKernelGGL((pairwise_l2_kernel<1, true>), grid, dim3(256), lds,
                           stream, x0, x1, x2, c0, c1, c2, xn, yn, out, m, n,
                           d, ldo, sqrt_out, rg);
      else
        hipLaunchKernelGGL((pairwise_l2_kernel<1>), grid, dim3(256), lds, stream,
                           x0, x1, x2, c0, c1, c2, xn, yn, out, m, n, d, ldo, sqrt_out, rg);
      break;
    case 2:
      if (bk32 && d % 32 == 0)
        hipLaunchKernelGGL((pairwise_l2_kernel<2, true>), grid, dim3(256), lds,
                           stream, x0, x1, x2, c0, c1, c2, xn, yn, out, m, n,
                           d, ldo, sqrt_out, rg);
      else
        hipLaunchKernelGGL((pairwise_l2_kernel<2>), grid, dim3(256), lds, stream,
                           x0, x1, x2, c0, c1, c2, xn, yn, out, m, n, d, ldo, sqrt_out, rg);
      break;
    case 3: {
      static bool attr_set = false;
      if (!attr_set) {
        HIP_CHECK(hipFuncSetAttribute((const void*)&pairwise_l2_kernel<3>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      96 * 1024));
        attr_set = true;
      }
      hipLaunchKernelGGL((pairwise_l2_kernel<3>), grid, dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, yn, out, m, n, d, ldo, sqrt_out, rg);
      break;
    }
    default:
      throw std::runtime_error("pairwise_l2_mfma: nslice must be 1, 2 or 3");
  }
}

}  // namespace raft_amd

namespace raft_amd {

// ---------------------------------------------------------------------------
// 256x256-tile variant: 512 threads (8 waves, 2x4 grid; per-wave 128x64
// fragment tile). Quarters the workgroup count of the 128x128 kernel —
// at small d (e.g. 128: only 2 K-steps) the per-workgroup prologue/epilogue
// is a first-order cost. LDS = NSLICE * 64 KiB.
// ---------------------------------------------------------------------------

template <int ROWS, int BLOCK>
__device__ __forceinline__ void pw_stage_rows(const __bf16* __restrict__ g,
                                              __bf16* lds, long long row0,
                                              long long k0, long long ld,
                                              long long max_row) {
  const int t = threadIdx.x;
  const int w = t / RAFT_AMD_WAVE;
  constexpr int ROUNDS = ROWS * 128 / (BLOCK * 16);
#pragma unroll
  for (int j = 0; j < ROUNDS; j++) {
    const int o = j * BLOCK * 16 + t * 16;
    const int o_src = mfma_swz(o);
    long long r = row0 + (o_src >> 7);
    if (r > max_row) r = max_row;
    const long long goff = r * ld + k0 + ((o_src & 127) >> 1);
    __bf16* lbase = lds + (j * BLOCK * 16 + w * 1024) / 2;
    GLOAD_LDS(g + goff, lbase);
  }
}

template <int NSLICE>
__launch_bounds__(512, 1)
__global__ void pairwise_l2_256_kernel(const __bf16* __restrict__ x0,
                                       const __bf16* __restrict__ x1,
                                       const __bf16* __restrict__ x2,
                                       const __bf16* __restrict__ c0,
                                       const __bf16* __restrict__ c1,
                                       const __bf16* __restrict__ c2,
                                       const float* __restrict__ xn,
                                       const float* __restrict__ yn,
                                       float* __restrict__ out,
                                       long long m, long long n, int d,
                                       long long ldo, int sqrt_out, int rg) {
  constexpr int BLOCK = 512;
  extern __shared__ __bf16 smem[];

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 2, wc = w & 3;  // 2x4: wave tile 128 rows x 64 cols
  long long rt_, ct_;
  xcd_supertile_decode(rg, &rt_, &ct_);
  if (rt_ * 256 >= m || ct_ * 256 >= n) return;  // super-tile ragged edge
  const long long row0 = rt_ * 256;
  const long long col0 = ct_ * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int a = 0; a < 8; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  if constexpr (NSLICE <= 2) {
    // BK=32 product-phase counted-vmcnt K-loop (mfma_common.h): the old
    // per-K-step `vmcnt(0)+barrier` full drain ran at 1 block/CU with
    // nothing to hide it — at d=128 (2 K-steps/tile) the drains dominated.
    Mfma256BK32 st;
    mfma256_bk32_setup(st, row0, col0, d, m - 1, n - 1, wr, wc, lane);
    mfma256_bk32_kloop<NSLICE>(x0, x1, c0, c1, smem, st, acc, d / 32);
  }
  static_assert(NSLICE <= 2, "256^2 pairwise kernel: nslice 3 routes to the 128^2 kernel");

  if (row0 + 256 <= m && col0 + 256 <= n && (ldo & 3) == 0) {
    // interior tile: stage each 128-row half through padded LDS
    // ([128][260] fp32, 130 KiB) and dump as coalesced f32x4 rows —
    // the per-element path issues 32 scalar 4 B stores/thread with 64-bit
    // row*ldo address math each. sqrt_out bit 1 selects nontemporal vs
    // cached stores (A/B: write-combine behavior differs).
    float* tile = reinterpret_cast<float*>(smem);  // [128][260]
    const int tid = threadIdx.x;
    const int tr = tid >> 6;                // 0..7
    const int tc = (tid & 63) * 4;          // 0..252
    const bool nt = (sqrt_out & 2) != 0;
    const bool do_sqrt = (sqrt_out & 1) != 0;
#pragma unroll
    for (int half = 0; half < 2; half++) {
      __syncthreads();
      if (wr == half) {
#pragma unroll
        for (int fr = 0; fr < 8; fr++) {
#pragma unroll
          for (int reg = 0; reg < 4; reg++) {
            const int rl = fr * 16 + (lane >> 4) * 4 + reg;
            const float xv = xn[row0 + half * 128 + rl];
#pragma unroll
            for (int fc = 0; fc < 4; fc++) {
              const int cl = wc * 64 + fc * 16 + (lane & 15);
              float v =
                  fmaxf(xv + yn[col0 + cl] - 2.f * acc[fr][fc][reg], 0.f);
              if (do_sqrt) v = sqrtf(v);
              tile[rl * 260 + cl] = v;
            }
          }
        }
      }
      __syncthreads();
      float* dst0 = out + (row0 + half * 128 + tr) * ldo + col0 + tc;
      const float* src0 = tile + tr * 260 + tc;
#pragma unroll
      for (int rnd = 0; rnd < 16; rnd++) {
        const f32x4 v4 = *reinterpret_cast<const f32x4*>(src0 + rnd * 8 * 260);
        f32x4* dp = reinterpret_cast<f32x4*>(dst0 + (long long)rnd * 8 * ldo);
        if (nt)
          __builtin_nontemporal_store(v4, dp);
        else
          *dp = v4;
      }
    }
    return;
  }
#pragma unroll
  for (int fr = 0; fr < 8; fr++) {
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      const long long row = row0 + wr * 128 + fr * 16 + (lane >> 4) * 4 + reg;
      if (row >= m) continue;
      const float xv = xn[row];
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const long long col = col0 + wc * 64 + fc * 16 + (lane & 15);
        if (col < n) {
          float v = fmaxf(xv + yn[col] - 2.f * acc[fr][fc][reg], 0.f);
          if (sqrt_out & 1) v = sqrtf(v);
          // streaming output (never re-read): non-temporal keeps the L2
          // clear for the operand tiles
          __builtin_nontemporal_store(v, &out[row * ldo + col]);
        }
      }
    }
  }
}

void launch_pairwise_l2_mfma256(const void** xsl, const void** csl, const float* xn,
                                const float* yn, float* out, long long m, long long n,
                                int d, long long ldo, int nslice, bool sqrt_out,
                                hipStream_t stream) {
  const int rg = (int)((m + 2047) / 2048);       // ceil(R/8), R=ceil(m/256)
  const int cg = (int)((n + 2047) / 2048);
  dim3 grid((unsigned)((long long)rg * cg * 64));
  // K-loop: 8 x 16 KiB BK=32 dbuf regions (128 KiB); epilogue staging:
  // [128][260] fp32 (130 KiB) — allocate the max
  const size_t lds_k = 8 * 16384;
  const size_t lds_epi = 128 * 260 * 4;
  const size_t lds = lds_k > lds_epi ? lds_k : lds_epi;
  static const bool nt_store = [] {
    const char* e = getenv("RAFT_AMD_PW_NT");
    return e && e[0] == '1';  // default: cached (write-combining via L2)
  }();
  const int so_bits = (sqrt_out ? 1 : 0) | (nt_store ? 2 : 0);
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  auto set_attr = [](const void* f) {
    HIP_CHECK(hipFuncSetAttribute(f, hipFuncAttributeMaxDynamicSharedMemorySize,
                                  160 * 1024));
  };
  switch (nslice) {
    case 1: {
      static bool a1 = (set_attr((const void*)&pairwise_l2_256_kernel<1>), true);
      (void)a1;
      hipLaunchKernelGGL((pairwise_l2_256_kernel<1>), grid, dim3(512), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, yn, out, m, n, d, ldo, so_bits, rg);
      break;
    }
    case 2: {
      static bool a2 = (set_attr((const void*)&pairwise_l2_256_kernel<2>), true);
      (void)a2;
      hipLaunchKernelGGL((pairwise_l2_256_kernel<2>), grid, dim3(512), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, yn, out, m, n, d, ldo, so_bits, rg);
      break;
    }
    default:
      throw std::runtime_error("pairwise 256^2 kernel: nslice must be 1 or 2");
  }
}

}  // namespace raft_amd

namespace raft_amd {

// 256x256-tile filtered emission (see pairwise_l2_256_kernel geometry +
// pairwise_l2_filter_kernel semantics)
template <int NSLICE>
__launch_bounds__(512, 2)
__global__ void pairwise_l2_filter256_kernel(const __bf16* __restrict__ x0,
                                             const __bf16* __restrict__ x1,
                                             const __bf16* __restrict__ x2,
                                             const __bf16* __restrict__ c0,
                                             const __bf16* __restrict__ c1,
                                             const __bf16* __restrict__ c2,
                                             const float* __restrict__ xn,
                                             const float* __restrict__ yn,
                                             const float* __restrict__ thr,
                                             float* __restrict__ out_d,
                                             int* __restrict__ out_i,
                                             int* __restrict__ cnt, int cap,
                                             long long col_offset, long long m,
                                             long long n, int d, int rg) {
  constexpr int BLOCK = 512;
  extern __shared__ __bf16 smem[];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 16384;
    cs[s] = smem + NSLICE * 16384 + s * 16384;
  }
  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 2, wc = w & 3;
  long long rt_, ct_;
  xcd_supertile_decode(rg, &rt_, &ct_);
  if (rt_ * 256 >= m || ct_ * 256 >= n) return;  // super-tile ragged edge
  const long long row0 = rt_ * 256;
  const long long col0 = ct_ * 256;

  f32x4 acc[8][4];
#pragma unroll
  for (int a = 0; a < 8; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  if constexpr (NSLICE <= 2) {
    // counted-vmcnt BK=32 dbuf K-loop (mfma_common.h): the old full
    // per-K-step drain ran at 1 block/CU with nothing to hide it — this
    // kernel has no tile writes, so the K-loop IS the kernel
    // (RAFT_AMD_KNN_DRAIN=1 in the launcher restores the old loop for A/B)
    Mfma256BK32 st;
    mfma256_bk32_setup<256>(st, row0, col0, d, m - 1, n - 1, wr, wc, lane);
    mfma256_bk32_kloop<NSLICE, 256>(x0, x1, c0, c1, smem, st, acc, d / 32);
  } else {
    const int k_tiles = d / 64;
    for (int kt = 0; kt < k_tiles; kt++) {
#pragma unroll
      for (int s = 0; s < NSLICE; s++) {
        pw_stage_rows<256, BLOCK>(xg[s], xs[s], row0, (long long)kt * 64, d, m - 1);
        pw_stage_rows<256, BLOCK>(cg[s], cs[s], col0, (long long)kt * 64, d, n - 1);
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
#pragma unroll
      for (int kf = 0; kf < 2; kf++) {
        bf16x8 a_frag[NSLICE][8], b_frag[NSLICE][4];
        const int kbyte = (kf * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
        for (int fr = 0; fr < 8; fr++) {
          const int r = wr * 128 + fr * 16 + (lane & 15);
          const int byte = mfma_swz(r * 128 + kbyte);
#pragma unroll
          for (int s = 0; s < NSLICE; s++)
            a_frag[s][fr] = *reinterpret_cast<const bf16x8*>((const char*)xs[s] + byte);
        }
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int c = wc * 64 + fc * 16 + (lane & 15);
          const int byte = mfma_swz(c * 128 + kbyte);
#pragma unroll
          for (int s = 0; s < NSLICE; s++)
            b_frag[s][fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[s] + byte);
        }
#pragma unroll
        for (int fr = 0; fr < 8; fr++)
#pragma unroll
          for (int fc = 0; fc < 4; fc++) {
#pragma unroll
            for (int p = 0; p < mfma_n_products<NSLICE>(); p++) {
              acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  a_frag[MFMA_PROD_A[p]][fr], b_frag[MFMA_PROD_B[p]][fc],
                  acc[fr][fc], 0, 0, 0);
            }
          }
      }
      __syncthreads();
    }
  }

  const long long colb = col0 + wc * 64 + (lane & 15);
  float yn_r[4];
#pragma unroll
  for (int fc = 0; fc < 4; fc++) {
    const long long col = colb + fc * 16;
    yn_r[fc] = col < n ? yn[col] : INFINITY;
  }
#pragma unroll
  for (int fr = 0; fr < 8; fr++) {
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      const long long row = row0 + wr * 128 + fr * 16 + (lane >> 4) * 4 + reg;
      if (row >= m) continue;
      const float xv = xn[row];
      const float t = fmaxf(thr[row], 0.f);
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const float sd = xv + yn_r[fc] - 2.f * acc[fr][fc][reg];
        if (sd <= t) {
          const int pos = atomicAdd(&cnt[row], 1);
          if (pos < cap) {
            out_d[row * cap + pos] = fmaxf(sd, 0.f);
            out_i[row * cap + pos] = (int)(colb + fc * 16 + col_offset);
          }
        }
      }
    }
  }
}

void launch_pairwise_l2_filter256(const void** xsl, const void** csl, const float* xn,
                                  const float* yn, const float* thr, float* out_d,
                                  int* out_i, int* cnt, int cap, long long col_offset,
                                  long long m, long long n, int d, int nslice,
                                  hipStream_t stream) {
  const int rg = (int)((m + 2047) / 2048);       // ceil(R/8), R=ceil(m/256)
  const int cg = (int)((n + 2047) / 2048);
  dim3 grid((unsigned)((long long)rg * cg * 64));
  // nslice<=2: BK=32 counted-vmcnt loop needs the 8 x 16 KiB dbuf regions
  const size_t lds = nslice <= 2 ? (size_t)(8 * 16384)
                                 : (size_t)nslice * 2 * 16384 * sizeof(__bf16);
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  auto set_attr = [](const void* f) {
    HIP_CHECK(hipFuncSetAttribute(f, hipFuncAttributeMaxDynamicSharedMemorySize,
                                  160 * 1024));
  };
  if (nslice == 1) {
    static bool a1 = (set_attr((const void*)&pairwise_l2_filter256_kernel<1>), true);
    (void)a1;
    hipLaunchKernelGGL((pairwise_l2_filter256_kernel<1>), grid, dim3(512), lds, stream,
                       x0, x1, x2, c0, c1, c2, xn, yn, thr, out_d, out_i, cnt, cap,
                       col_offset, m, n, d, rg);
  } else if (nslice == 2) {
    static bool a2 = (set_attr((const void*)&pairwise_l2_filter256_kernel<2>), true);
    (void)a2;
    hipLaunchKernelGGL((pairwise_l2_filter256_kernel<2>), grid, dim3(512), lds, stream,
                       x0, x1, x2, c0, c1, c2, xn, yn, thr, out_d, out_i, cnt, cap,
                       col_offset, m, n, d, rg);
  } else {
    static bool a3 = (set_attr((const void*)&pairwise_l2_filter256_kernel<3>), true);
    (void)a3;
    hipLaunchKernelGGL((pairwise_l2_filter256_kernel<3>), grid, dim3(512), lds, stream,
                       x0, x1, x2, c0, c1, c2, xn, yn, thr, out_d, out_i, cnt, cap,
                       col_offset, m, n, d, rg);
  }
}

}  // namespace raft_amd
