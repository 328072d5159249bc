// Shared machinery for split-bf16 MFMA contraction kernels (gfx950).
// Reference parity: the contractions engine role (raft/linalg/contractions.cuh
// Contractions_NT, KernelPolicy) re-derived for 64-wide wavefronts and MFMA.
// See fused_l2nn.hip header comment for the design rationale (tile geometry,
// XOR swizzle + global_load_lds both-sides rule, slice-product emulation).
#pragma once

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GLOAD_LDS(gp, lp)                                                      \
  __builtin_amdgcn_global_load_lds(                                           \
      (const __attribute__((address_space(1))) void*)(gp),                    \
      (__attribute__((address_space(3))) void*)(lp), 16, 0, 0)

// swizzle: flip byte-offset bit4 by row bits (rows are 128 B = 64 bf16)
__device__ __forceinline__ int mfma_swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

// stage a [128][64] bf16 tile from row-major global (leading dim ld elements)
// into a swizzled LDS tile (16 KiB). 256 threads, 4 gload rounds.
__device__ __forceinline__ void mfma_stage_tile128(const __bf16* __restrict__ g,
                                                   __bf16* lds, long long row0,
                                                   long long k0, long long ld,
                                                   long long max_row) {
  const int t = threadIdx.x;
  const int w = t / RAFT_AMD_WAVE;
#pragma unroll
  for (int j = 0; j < 4; j++) {
    const int o = j * 4096 + t * 16;  // linear dest byte
    const int o_src = mfma_swz(o);    // fetch what belongs here
    long long r = row0 + (o_src >> 7);
    if (r > max_row) r = max_row;
    const long long goff = r * ld + k0 + ((o_src & 127) >> 1);
    __bf16* lbase = lds + (j * 4096 + w * 1024) / 2;
    GLOAD_LDS(g + goff, lbase);
  }
}

// XCD-contiguous 8x8 super-tiled (row-tile, col-tile) decode for 2D tile
// grids (guide T1 + super-tiling): the bijective remap gives each XCD a
// contiguous slot range, and consecutive 64-slot windows decode to an
// 8x8 super-tile — the window's 8 X panels + 8 C panels (512 KiB at
// 128-tiles) are read into that XCD's 4 MiB L2 once and reused 8x each.
// Launch with grid = rg*cg*64 (rg=ceil(R/8), cg=ceil(C/8)); callers must
// early-return when rt/ct land past R/C (grid inflation on ragged edges).
// rg < 0 selects COLUMN-major slot order within the 8x8 window (consecutive
// slots walk adjacent col-tiles of one row-tile — adjacent 512 B output
// segments, better DRAM page locality for tile-writing kernels); rg > 0 is
// the original row-major order. L2 panel reuse is symmetric under the swap.
__device__ __forceinline__ void xcd_supertile_decode(int rg, long long* rt,
                                                     long long* ct) {
  const bool cm = rg < 0;
  if (cm) rg = -rg;
  const int nwg = gridDim.x;
  const int bid = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, slot = bid >> 3;
  const int t = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  const int g = t >> 6, w = t & 63;
  const int wr = cm ? (w >> 3) : (w & 7);
  const int wc = cm ? (w & 7) : (w >> 3);
  *rt = (long long)(g % rg) * 8 + wr;
  *ct = (long long)(g / rg) * 8 + wc;
}

template <int NSLICE>
__device__ __forceinline__ constexpr int mfma_n_products() {
  return NSLICE == 1 ? 1 : (NSLICE == 2 ? 3 : 6);
}

__device__ constexpr int MFMA_PROD_A[6] = {0, 0, 1, 1, 0, 2};
__device__ constexpr int MFMA_PROD_B[6] = {0, 1, 0, 1, 2, 0};

// Run the K loop for one 128x128 tile pair: stages slices, MFMAs into acc.
// acc is the per-wave [4][4] fragment grid (wave (wr,wc) of a 2x2 wave grid).
// Staging source addresses are hoisted: the per-thread (row, kcol) of each
// gload round is K-invariant, so the 64-bit row*ld multiplies run once per
// tile pair instead of once per K-step.
template <int NSLICE>
__device__ __forceinline__ void mfma_tile_kloop(
    const __bf16* const (&xg)[3], const __bf16* const (&cg)[3],
    __bf16* (&xs)[NSLICE], __bf16* (&cs)[NSLICE],
    f32x4 (&acc)[4][4], long long row0, long long col0, int d,
    long long m_max, long long n_max, int wr, int wc, int lane) {
  const int k_tiles = d / 64;
  const int t = threadIdx.x;
  const int wv = t / RAFT_AMD_WAVE;
  long long bx[4], bc[4];
  int ldst[4];
#pragma unroll
  for (int j = 0; j < 4; j++) {
    const int o = j * 4096 + t * 16;
    const int o_src = mfma_swz(o);
    const int r = o_src >> 7;
    const int k = (o_src & 127) >> 1;
    long long rx = row0 + r;
    if (rx > m_max) rx = m_max;
    bx[j] = rx * (long long)d + k;
    long long rc = col0 + r;
    if (rc > n_max) rc = n_max;
    bc[j] = rc * (long long)d + k;
    ldst[j] = (j * 4096 + wv * 1024) / 2;
  }
  for (int kt = 0; kt < k_tiles; kt++) {
    const long long koff = (long long)kt * 64;
#pragma unroll
    for (int s = 0; s < NSLICE; s++) {
#pragma unroll
      for (int j = 0; j < 4; j++) {
        GLOAD_LDS(xg[s] + bx[j] + koff, xs[s] + ldst[j]);
        GLOAD_LDS(cg[s] + bc[j] + koff, cs[s] + ldst[j]);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

#pragma unroll
    for (int kf = 0; kf < 2; kf++) {
      bf16x8 a_frag[NSLICE][4], b_frag[NSLICE][4];
#pragma unroll
      for (int fr = 0; fr < 4; fr++) {
        const int r = wr * 64 + fr * 16 + (lane & 15);
        const int byte = mfma_swz(r * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
#pragma unroll
        for (int s = 0; s < NSLICE; s++)
          a_frag[s][fr] = *reinterpret_cast<const bf16x8*>((const char*)xs[s] + byte);
      }
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const int c = wc * 64 + fc * 16 + (lane & 15);
        const int byte = mfma_swz(c * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
#pragma unroll
        for (int s = 0; s < NSLICE; s++)
          b_frag[s][fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[s] + byte);
      }
#pragma unroll
      for (int fr = 0; fr < 4; fr++)
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

// ---------------------------------------------------------------------------
// 256x256-tile BK=32 product-phase counted-vmcnt K-loop (8 waves, 2x4 grid,
// per-wave 128x64 output, acc[8][4]). All slice tiles of one K-chunk are
// resident (A0,A1,B0,B1 = 4 x 16 KiB), double-buffered in 128 KiB LDS; the
// NSLICE=2 split-bf16 products run as 3 phases from ONE staging with
// counted `s_waitcnt vmcnt(N)` (no full drains in the main loop); NSLICE=1
// uses the 8 regions as a 4-deep ring staging 2 chunks ahead. See
// fused_l2nn_256.hip header + profiles/pmc_l2nn_256_ab.txt for the measured
// design history (swizzle, addrspace and spill lessons).
//
// LDS region layout (16 KiB = 8192 bf16 each):
//   A_ELE(buf, s) = (buf*4 + s)     * 8192
//   B_ELE(buf, s) = (buf*4 + 2 + s) * 8192
// ---------------------------------------------------------------------------

// region size in bf16 elements for a [ROWS][32] tile (ROWS = 256 or 128)
#define RAFT_MFMA256_REG(ROWS) ((ROWS) * 32)
#define RAFT_MFMA256_A_ELE(buf, s) (((buf) * 4 + (s)) * RAFT_MFMA256_REG(ROWS))
#define RAFT_MFMA256_B_ELE(buf, s) (((buf) * 4 + 2 + (s)) * RAFT_MFMA256_REG(ROWS))

struct Mfma256BK32 {
  long long bx[2], bc[2];  // per-thread global staging offsets (2 rounds)
  int ldst[2];             // LDS dest offsets (bf16 elements)
  int a_off[8], b_off[4];  // hoisted ds-read byte offsets (a_off[ROWS/32])
};

// Balanced ADD-rotation swizzle for 64 B LDS rows: row r's k-slot s (16 B
// units) stored at slot (s + (r>>1)) & 3 — a 16-row b128 column read covers
// every 128 B bank window exactly 2x (the minimum; the XOR swizzle measures
// 1.0 conflicts/MFMA here).
template <int ROWS = 256>
__device__ __forceinline__ void mfma256_bk32_setup(
    Mfma256BK32& st, long long row0, long long col0, int d, long long m_max,
    long long n_max, int wm, int wn, int lane) {
  constexpr int BLOCK = ROWS * 2;   // threads: 512 (256^2) / 256 (128^2)
  const int tid = threadIdx.x;
  const int w = tid >> 6;
#pragma unroll
  for (int j = 0; j < 2; j++) {
    const int o = j * BLOCK * 16 + tid * 16;   // 16-aligned linear dest byte
    const int rr = o >> 6;               // dest row 0..255
    const int sd = (o >> 4) & 3;         // dest slot
    const int kk = (((sd - (rr >> 1)) & 3) << 3);  // source k (bf16 elems)
    long long rx = row0 + rr;
    if (rx > m_max) rx = m_max;
    st.bx[j] = rx * (long long)d + kk;
    long long rc = col0 + rr;
    if (rc > n_max) rc = n_max;
    st.bc[j] = rc * (long long)d + kk;
    st.ldst[j] = (j * BLOCK * 16 + w * 1024) / 2;
  }
  const int ks = lane >> 4;            // K slot 0..3 (8 bf16 each)
#pragma unroll
  for (int fr = 0; fr < ROWS / 32; fr++) {
    const int rr = wm * (ROWS / 2) + fr * 16 + (lane & 15);
    st.a_off[fr] = rr * 64 + (((ks + (rr >> 1)) & 3) << 4);
  }
#pragma unroll
  for (int fc = 0; fc < 4; fc++) {
    const int cc = wn * 64 + fc * 16 + (lane & 15);
    st.b_off[fc] = cc * 64 + (((ks + (cc >> 1)) & 3) << 4);
  }
}

template <int NSLICE, int ROWS = 256>
__device__ __forceinline__ void mfma256_bk32_kloop(
    const __bf16* __restrict__ x0, const __bf16* __restrict__ x1,
    const __bf16* __restrict__ c0, const __bf16* __restrict__ c1,
    __bf16* smem, const Mfma256BK32& st, f32x4 (&acc)[ROWS / 32][4],
    int kt_tiles) {
  static_assert(NSLICE <= 2, "mfma256_bk32_kloop: NSLICE 1 or 2");
  constexpr int FR = ROWS / 32;
#define M256_GA(slice, buf, koff)                                              \
  do {                                                                         \
    GLOAD_LDS((slice == 0 ? x0 : x1) + st.bx[0] + (koff),                      \
              smem + RAFT_MFMA256_A_ELE(buf, slice) + st.ldst[0]);             \
    GLOAD_LDS((slice == 0 ? x0 : x1) + st.bx[1] + (koff),                      \
              smem + RAFT_MFMA256_A_ELE(buf, slice) + st.ldst[1]);             \
  } while (0)
#define M256_GB(slice, buf, koff)                                              \
  do {                                                                         \
    GLOAD_LDS((slice == 0 ? c0 : c1) + st.bc[0] + (koff),                      \
              smem + RAFT_MFMA256_B_ELE(buf, slice) + st.ldst[0]);             \
    GLOAD_LDS((slice == 0 ? c0 : c1) + st.bc[1] + (koff),                      \
              smem + RAFT_MFMA256_B_ELE(buf, slice) + st.ldst[1]);             \
  } while (0)
  const char* lds_base = reinterpret_cast<const char*>(smem);
  auto ds_b = [&](int buf_ele, bf16x8(&b_frag)[4]) {
    const char* base = lds_base + buf_ele * 2;
#pragma unroll
    for (int fc = 0; fc < 4; fc++)
      b_frag[fc] = *reinterpret_cast<const bf16x8*>(base + st.b_off[fc]);
  };
  // A fragments stream through a 2-row register window (a full a_frag[8]
  // per slice pushes peak pressure past 256 VGPRs -> K-loop scratch spills,
  // whose vmem ops also corrupt the counted vmcnt arithmetic)
  auto mfma32 = [&](int buf_a_ele, const bf16x8(&b_frag)[4]) {
    const char* abase = lds_base + buf_a_ele * 2;
#pragma unroll
    for (int qd = 0; qd < FR / 2; qd++) {
      bf16x8 a2[2];
#pragma unroll
      for (int fi = 0; fi < 2; fi++)
        a2[fi] =
            *reinterpret_cast<const bf16x8*>(abase + st.a_off[qd * 2 + fi]);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fi = 0; fi < 2; fi++)
#pragma unroll
        for (int fc = 0; fc < 4; fc++)
          acc[qd * 2 + fi][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a2[fi], b_frag[fc], acc[qd * 2 + fi][fc], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  if constexpr (NSLICE == 1) {
    // 4-deep (A,B) ring staging TWO K-chunks ahead: one barrier + one
    // counted wait per chunk; a chunk's loads get 2 chunks (64 MFMA) of
    // flight time before their wait.
#define M256_RING_A(r) RAFT_MFMA256_A_ELE((r) >> 1, (r)&1)
#define M256_RING_B(r) RAFT_MFMA256_B_ELE((r) >> 1, (r)&1)
    M256_GB(0, 0, 0);                      // kt0 -> ring 0
    M256_GA(0, 0, 0);
#pragma unroll
    for (int j = 0; j < 2; j++) {          // kt1 -> ring 1
      GLOAD_LDS(c0 + st.bc[j] + 32, smem + M256_RING_B(1) + st.ldst[j]);
      GLOAD_LDS(x0 + st.bx[j] + 32, smem + M256_RING_A(1) + st.ldst[j]);
    }
    for (int kt = 0; kt < kt_tiles; kt++) {
      const int cur = kt & 3;
      if (kt + 2 < kt_tiles) {
        const long long koff = (long long)(kt + 2) * 32;
        const int nxt = (kt + 2) & 3;
#pragma unroll
        for (int j = 0; j < 2; j++) {
          GLOAD_LDS(c0 + st.bc[j] + koff, smem + M256_RING_B(nxt) + st.ldst[j]);
          GLOAD_LDS(x0 + st.bx[j] + koff, smem + M256_RING_A(nxt) + st.ldst[j]);
        }
        asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
      } else if (kt + 1 < kt_tiles) {
        asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
      }
      bf16x8 b_frag[4];
      ds_b(M256_RING_B(cur), b_frag);
      mfma32(M256_RING_A(cur), b_frag);
    }
#undef M256_RING_A
#undef M256_RING_B
  } else {
    // split-bf16: 3 product-phases per K-chunk from ONE staging (96 MFMA
    // per 64 KiB staged). Per phase: issue kt+1's gloads, ONE counted wait
    // + ONE barrier, ds-read, MFMA; the phase-0 wait is loop-carried.
    // Issue order per chunk: [B0r0 B0r1 A0r0] [A0r1 B1r0 B1r1] [A1r0 A1r1].
    M256_GB(0, 0, 0);
    M256_GA(0, 0, 0);
    M256_GB(1, 0, 0);
    M256_GA(1, 0, 0);
    // phase 0 needs B0+A0 (leave B1,A1 in flight)
    asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
    for (int kt = 0; kt < kt_tiles; kt++) {
      const int cur = kt & 1;
      const bool more = kt + 1 < kt_tiles;
      const long long koff = (long long)(kt + 1) * 32;
      bf16x8 b0[4], b1[4];
      // ---- phase 0: p00 = A0 x B0 (wait carried from prev phase 2) -------
      if (more) {
        M256_GB(0, cur ^ 1, koff);
        GLOAD_LDS(x0 + st.bx[0] + koff,
                  smem + RAFT_MFMA256_A_ELE(cur ^ 1, 0) + st.ldst[0]);
      }
      ds_b(RAFT_MFMA256_B_ELE(cur, 0), b0);
      mfma32(RAFT_MFMA256_A_ELE(cur, 0), b0);
      // ---- phase 1: p01 = A0 x B1 (first read of B1(kt)) -----------------
      if (more) {
        GLOAD_LDS(x0 + st.bx[1] + koff,
                  smem + RAFT_MFMA256_A_ELE(cur ^ 1, 0) + st.ldst[1]);
        M256_GB(1, cur ^ 1, koff);
        // in flight: B1,A1(kt) + 6(kt+1); retire B1(kt)
        asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(2)\n\ts_barrier" ::: "memory");
      }
      ds_b(RAFT_MFMA256_B_ELE(cur, 1), b1);
      mfma32(RAFT_MFMA256_A_ELE(cur, 0), b1);
      // ---- phase 2: p10 = A1 x B0 (first read of A1(kt)) -----------------
      if (more) {
        M256_GA(1, cur ^ 1, koff);
        // in flight: A1(kt) + 8(kt+1); retire A1(kt)
        asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
      }
      mfma32(RAFT_MFMA256_A_ELE(cur, 1), b0);
      // next chunk's phase 0 reads B0,A0(kt+1): retire the oldest 4
      if (more) {
        asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
      } else {
        asm volatile("s_barrier" ::: "memory");
      }
    }
  }
#undef M256_GA
#undef M256_GB
}

// ---------------------------------------------------------------------------
// A-direct K-loop: X fragments load straight global->register (no LDS leg).
// Within a block each X element is read by exactly ONE wave exactly once, so
// LDS-staging X is a pure write+read round-trip — the 2-block/CU 128^2
// engine is LDS-write bound (128 KiB staged per CU per K-tile vs 768 MFMA
// ~= 1024 vs 768 cycles). Dropping the X leg halves LDS writes, halves the
// per-K-step drain (8 gloads instead of 16) and shrinks LDS to C-only
// (NSLICE*16 KiB). X reads hit the XCD-local L2 (row-tile remap keeps the
// panel hot). C stays LDS-staged: its fragments are read by all 4 waves.
// ---------------------------------------------------------------------------
template <int NSLICE>
__device__ __forceinline__ void mfma_tile_kloop_ad(
    const __bf16* const (&xg)[3], const __bf16* const (&cg)[3],
    __bf16* (&cs)[NSLICE],
    f32x4 (&acc)[4][4], long long row0, long long col0, int d,
    long long m_max, long long n_max, int wr, int wc, int lane) {
  const int k_tiles = d / 64;
  const int t = threadIdx.x;
  const int wv = t / RAFT_AMD_WAVE;
  long long bc[4];
  int ldst[4];
#pragma unroll
  for (int j = 0; j < 4; j++) {
    const int o = j * 4096 + t * 16;
    const int o_src = mfma_swz(o);
    const int r = o_src >> 7;
    const int k = (o_src & 127) >> 1;
    long long rc = col0 + r;
    if (rc > n_max) rc = n_max;
    bc[j] = rc * (long long)d + k;
    ldst[j] = (j * 4096 + wv * 1024) / 2;
  }
  // hoisted per-fragment X row base offsets (global elements)
  long long ax[4];
#pragma unroll
  for (int fr = 0; fr < 4; fr++) {
    long long r = row0 + wr * 64 + fr * 16 + (lane & 15);
    if (r > m_max) r = m_max;
    ax[fr] = r * (long long)d + (lane >> 4) * 8;
  }
  for (int kt = 0; kt < k_tiles; kt++) {
    const long long koff = (long long)kt * 64;
#pragma unroll
    for (int s = 0; s < NSLICE; s++)
#pragma unroll
      for (int j = 0; j < 4; j++)
        GLOAD_LDS(cg[s] + bc[j] + koff, cs[s] + ldst[j]);
    // X fragments direct from global, issued before the drain so their
    // latency overlaps the C-staging wait
    bf16x8 a_frag[NSLICE][2][4];
#pragma unroll
    for (int s = 0; s < NSLICE; s++)
#pragma unroll
      for (int kf = 0; kf < 2; kf++)
#pragma unroll
        for (int fr = 0; fr < 4; fr++)
          a_frag[s][kf][fr] = *reinterpret_cast<const bf16x8*>(
              xg[s] + ax[fr] + koff + kf * 32);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
#pragma unroll
    for (int kf = 0; kf < 2; kf++) {
      bf16x8 b_frag[NSLICE][4];
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const int c = wc * 64 + fc * 16 + (lane & 15);
        const int byte = mfma_swz(c * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
#pragma unroll
        for (int s = 0; s < NSLICE; s++)
          b_frag[s][fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[s] + byte);
      }
#pragma unroll
      for (int fr = 0; fr < 4; fr++)
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
#pragma unroll
          for (int p = 0; p < mfma_n_products<NSLICE>(); p++) {
            acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag[MFMA_PROD_A[p]][kf][fr], b_frag[MFMA_PROD_B[p]][fc],
                acc[fr][fc], 0, 0, 0);
          }
        }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// 1-slice paired K-loop: stage TWO k-tiles (64 KiB LDS) under ONE
// vmcnt(0)+barrier drain. MEASURED LOSER on the kNN filter sweep
// (9873 -> 7409 q/s, d=128): halving the drain count does not pay for the
// doubled per-drain wait (16 outstanding gloads) — kept only as a recorded
// experiment, not routed from any kernel.
// ---------------------------------------------------------------------------
__device__ __forceinline__ void mfma_tile_kloop_1s_pair(
    const __bf16* __restrict__ xg, const __bf16* __restrict__ cg,
    __bf16* (&xs)[2], __bf16* (&cs)[2],
    f32x4 (&acc)[4][4], long long row0, long long col0, int d,
    long long m_max, long long n_max, int wr, int wc, int lane) {
  const int k_tiles = d / 64;
  const int t = threadIdx.x;
  const int wv = t / RAFT_AMD_WAVE;
  long long bx[4], bc[4];
  int ldst[4];
#pragma unroll
  for (int j = 0; j < 4; j++) {
    const int o = j * 4096 + t * 16;
    const int o_src = mfma_swz(o);
    const int r = o_src >> 7;
    const int k = (o_src & 127) >> 1;
    long long rx = row0 + r;
    if (rx > m_max) rx = m_max;
    bx[j] = rx * (long long)d + k;
    long long rc = col0 + r;
    if (rc > n_max) rc = n_max;
    bc[j] = rc * (long long)d + k;
    ldst[j] = (j * 4096 + wv * 1024) / 2;
  }
  for (int kt = 0; kt < k_tiles; kt += 2) {
    const int pair = (kt + 1 < k_tiles) ? 2 : 1;
#pragma unroll
    for (int h = 0; h < 2; h++) {
      if (h >= pair) break;
      const long long koff = (long long)(kt + h) * 64;
#pragma unroll
      for (int j = 0; j < 4; j++) {
        GLOAD_LDS(xg + bx[j] + koff, xs[h] + ldst[j]);
        GLOAD_LDS(cg + bc[j] + koff, cs[h] + ldst[j]);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
#pragma unroll
    for (int h = 0; h < 2; h++) {
      if (h >= pair) break;
#pragma unroll
      for (int kf = 0; kf < 2; kf++) {
        bf16x8 a_frag[4], b_frag[4];
#pragma unroll
        for (int fr = 0; fr < 4; fr++) {
          const int r = wr * 64 + fr * 16 + (lane & 15);
          const int byte = mfma_swz(r * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
          a_frag[fr] = *reinterpret_cast<const bf16x8*>((const char*)xs[h] + byte);
        }
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int c = wc * 64 + fc * 16 + (lane & 15);
          const int byte = mfma_swz(c * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
          b_frag[fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[h] + byte);
        }
#pragma unroll
        for (int fr = 0; fr < 4; fr++)
#pragma unroll
          for (int fc = 0; fc < 4; fc++)
            acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag[fr], b_frag[fc], acc[fr][fc], 0, 0, 0);
      }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Phased 2-slice K-loop (guide T3/T4: counted s_waitcnt vmcnt instead of a
// full drain). Load issue order per K-step: [X0 C0 | C1 | X1] (4 gload
// rounds each interleaved). The slice-product MFMAs are split into three
// phases by which tiles they touch:
//   phase A (vmcnt<=8):  p00 = X0*C0            — first 8 loads landed
//   phase B (vmcnt<=4):  p01 = X0*C1            — C1 landed
//   phase C (vmcnt==0):  p10 = X1*C0, p11=X1*C1 — all landed
// so the tail loads' HBM/L2 latency hides under phase A/B MFMA instead of a
// serial vmcnt(0) stall. Raw s_barrier via asm — the compiler's
// __syncthreads() would conservatively drain vmcnt(0) and defeat the
// counting. Cross-wave safety: every thread waits its OWN vmcnt before the
// barrier, so after it the corresponding loads of ALL threads have landed.
// ---------------------------------------------------------------------------

#define RAFT_AMD_WAIT_BARRIER(N)                                     \
  asm volatile("s_waitcnt vmcnt(" #N ")\n\ts_barrier" ::: "memory")

__device__ __forceinline__ void mfma_p2_frags(__bf16* ts, int wrc, int lane,
                                              bf16x8 (&frag)[2][4]) {
#pragma unroll
  for (int kf = 0; kf < 2; kf++)
#pragma unroll
    for (int f = 0; f < 4; f++) {
      const int r = wrc * 64 + f * 16 + (lane & 15);
      const int byte = mfma_swz(r * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
      frag[kf][f] = *reinterpret_cast<const bf16x8*>((const char*)ts + byte);
    }
}

__device__ __forceinline__ void mfma_p2_prod(const bf16x8 (&a)[2][4],
                                             const bf16x8 (&b)[2][4],
                                             f32x4 (&acc)[4][4]) {
#pragma unroll
  for (int kf = 0; kf < 2; kf++)
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int fc = 0; fc < 4; fc++)
        acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[kf][fr], b[kf][fc], acc[fr][fc], 0, 0, 0);
}

__device__ __forceinline__ void mfma_tile_kloop_p2(
    const __bf16* const (&xg)[3], const __bf16* const (&cg)[3],
    __bf16* (&xs)[2], __bf16* (&cs)[2],
    f32x4 (&acc)[4][4], long long row0, long long col0, int d,
    long long m_max, long long n_max, int wr, int wc, int lane) {
  const int k_tiles = d / 64;
  const int t = threadIdx.x;
  const int wv = t / RAFT_AMD_WAVE;
  long long bx[4], bc[4];
  int ldst[4];
#pragma unroll
  for (int j = 0; j < 4; j++) {
    const int o = j * 4096 + t * 16;
    const int o_src = mfma_swz(o);
    const int r = o_src >> 7;
    const int k = (o_src & 127) >> 1;
    long long rx = row0 + r;
    if (rx > m_max) rx = m_max;
    bx[j] = rx * (long long)d + k;
    long long rc = col0 + r;
    if (rc > n_max) rc = n_max;
    bc[j] = rc * (long long)d + k;
    ldst[j] = (j * 4096 + wv * 1024) / 2;
  }
  for (int kt = 0; kt < k_tiles; kt++) {
    const long long koff = (long long)kt * 64;
#pragma unroll
    for (int j = 0; j < 4; j++) {  // first 8: X0 + C0
      GLOAD_LDS(xg[0] + bx[j] + koff, xs[0] + ldst[j]);
      GLOAD_LDS(cg[0] + bc[j] + koff, cs[0] + ldst[j]);
    }
#pragma unroll
    for (int j = 0; j < 4; j++)    // next 4: C1
      GLOAD_LDS(cg[1] + bc[j] + koff, cs[1] + ldst[j]);
#pragma unroll
    for (int j = 0; j < 4; j++)    // last 4: X1
      GLOAD_LDS(xg[1] + bx[j] + koff, xs[1] + ldst[j]);

    bf16x8 a0[2][4], a1[2][4], b0[2][4], b1[2][4];
    RAFT_AMD_WAIT_BARRIER(8);      // X0,C0 landed (phase A)
    mfma_p2_frags(xs[0], wr, lane, a0);
    mfma_p2_frags(cs[0], wc, lane, b0);
    mfma_p2_prod(a0, b0, acc);     // p00
    RAFT_AMD_WAIT_BARRIER(4);      // C1 landed (phase B)
    mfma_p2_frags(cs[1], wc, lane, b1);
    mfma_p2_prod(a0, b1, acc);     // p01
    RAFT_AMD_WAIT_BARRIER(0);      // X1 landed (phase C)
    mfma_p2_frags(xs[1], wr, lane, a1);
    mfma_p2_prod(a1, b0, acc);     // p10
    mfma_p2_prod(a1, b1, acc);     // p11
    // all ds_reads must retire before the next K-step's stores
    asm volatile("s_waitcnt lgkmcnt(0)\n\ts_barrier" ::: "memory");
  }
}

// ---------------------------------------------------------------------------
// BK=32 double-buffered variant: 8 KiB tiles, 2-phase overlap (stage next
// tile while MFMAing the current one; single vmcnt(0)+barrier per K-step —
// the guide's minimum 2-phase recipe). LDS rows are 64 B, so the XOR swizzle
// spreads 4 rows over the 4 in-row 16 B slots (residual 4-way ds_read
// conflict ~1.6x on the LDS path — the price of fitting 2 blocks/CU).
// ---------------------------------------------------------------------------

__device__ __forceinline__ int mfma_swz32(int byte_off) {
  return byte_off ^ (((byte_off >> 6) & 3) << 4);
}

// stage a [128][32] bf16 tile (8 KiB) from row-major global
__device__ __forceinline__ void mfma_stage_tile128_bk32(const __bf16* __restrict__ g,
                                                        __bf16* lds, long long row0,
                                                        long long k0, long long ld,
                                                        long long max_row) {
  const int t = threadIdx.x;
  const int w = t / RAFT_AMD_WAVE;
#pragma unroll
  for (int j = 0; j < 2; j++) {
    const int o = j * 4096 + t * 16;
    const int o_src = mfma_swz32(o);
    long long r = row0 + (o_src >> 6);
    if (r > max_row) r = max_row;
    const long long goff = r * ld + k0 + ((o_src & 63) >> 1);
    __bf16* lbase = lds + (j * 4096 + w * 1024) / 2;
    GLOAD_LDS(g + goff, lbase);
  }
}

// Single-buffered BK=32 K-loop: 8 KiB tiles -> 32 KiB LDS total for 2 slices,
// enabling 4 blocks/CU (vs BK=64's 2) so more independent load-drains
// interleave on each SIMD. Pays the 64 B-row residual LDS read conflict.
template <int NSLICE>
__device__ __forceinline__ void mfma_tile_kloop_s32(
    const __bf16* const (&xg)[3], const __bf16* const (&cg)[3],
    __bf16* (&xs)[NSLICE], __bf16* (&cs)[NSLICE],
    f32x4 (&acc)[4][4], long long row0, long long col0, int d,
    long long m_max, long long n_max, int wr, int wc, int lane) {
  const int k_tiles = d / 32;
  const int t = threadIdx.x;
  const int wv = t / RAFT_AMD_WAVE;
  long long bx[2], bc[2];
  int ldst[2];
#pragma unroll
  for (int j = 0; j < 2; j++) {
    const int o = j * 4096 + t * 16;
    const int o_src = mfma_swz32(o);
    const int r = o_src >> 6;
    const int k = (o_src & 63) >> 1;
    long long rx = row0 + r;
    if (rx > m_max) rx = m_max;
    bx[j] = rx * (long long)d + k;
    long long rc = col0 + r;
    if (rc > n_max) rc = n_max;
    bc[j] = rc * (long long)d + k;
    ldst[j] = (j * 4096 + wv * 1024) / 2;
  }
  for (int kt = 0; kt < k_tiles; kt++) {
    const long long koff = (long long)kt * 32;
#pragma unroll
    for (int s = 0; s < NSLICE; s++)
#pragma unroll
      for (int j = 0; j < 2; j++) {
        GLOAD_LDS(xg[s] + bx[j] + koff, xs[s] + ldst[j]);
        GLOAD_LDS(cg[s] + bc[j] + koff, cs[s] + ldst[j]);
      }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    bf16x8 a_frag[NSLICE][4], b_frag[NSLICE][4];
    const int kbyte = (lane >> 4) * 16;
#pragma unroll
    for (int fr = 0; fr < 4; fr++) {
      const int r = wr * 64 + fr * 16 + (lane & 15);
      const int byte = mfma_swz32(r * 64 + kbyte);
#pragma unroll
      for (int s = 0; s < NSLICE; s++)
        a_frag[s][fr] = *reinterpret_cast<const bf16x8*>((const char*)xs[s] + byte);
    }
#pragma unroll
    for (int fc = 0; fc < 4; fc++) {
      const int c = wc * 64 + fc * 16 + (lane & 15);
      const int byte = mfma_swz32(c * 64 + kbyte);
#pragma unroll
      for (int s = 0; s < NSLICE; s++)
        b_frag[s][fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[s] + byte);
    }
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
#pragma unroll
        for (int p = 0; p < mfma_n_products<NSLICE>(); p++) {
          acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[MFMA_PROD_A[p]][fr], b_frag[MFMA_PROD_B[p]][fc],
              acc[fr][fc], 0, 0, 0);
        }
      }
    __syncthreads();
  }
}

template <int NSLICE>
__device__ __forceinline__ void mfma_tile_kloop_db32(
    const __bf16* const (&xg)[3], const __bf16* const (&cg)[3],
    __bf16* (&xs)[2][NSLICE], __bf16* (&cs)[2][NSLICE],
    f32x4 (&acc)[4][4], long long row0, long long col0, int d,
    long long m_max, long long n_max, int wr, int wc, int lane) {
  const int k_tiles = d / 32;
  auto stage = [&](int buf, int kt) {
#pragma unroll
    for (int s = 0; s < NSLICE; s++) {
      mfma_stage_tile128_bk32(xg[s], xs[buf][s], row0, (long long)kt * 32, d, m_max);
      mfma_stage_tile128_bk32(cg[s], cs[buf][s], col0, (long long)kt * 32, d, n_max);
    }
  };
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  int cur = 0;
  for (int kt = 0; kt < k_tiles; kt++) {
    if (kt + 1 < k_tiles) stage(cur ^ 1, kt + 1);  // overlap next-tile loads
    bf16x8 a_frag[NSLICE][4], b_frag[NSLICE][4];
    const int kbyte = (lane >> 4) * 16;  // (lane>>4)*8 bf16
#pragma unroll
    for (int fr = 0; fr < 4; fr++) {
      const int r = wr * 64 + fr * 16 + (lane & 15);
      const int byte = mfma_swz32(r * 64 + kbyte);
#pragma unroll
      for (int s = 0; s < NSLICE; s++)
        a_frag[s][fr] = *reinterpret_cast<const bf16x8*>((const char*)xs[cur][s] + byte);
    }
#pragma unroll
    for (int fc = 0; fc < 4; fc++) {
      const int c = wc * 64 + fc * 16 + (lane & 15);
      const int byte = mfma_swz32(c * 64 + kbyte);
#pragma unroll
      for (int s = 0; s < NSLICE; s++)
        b_frag[s][fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[cur][s] + byte);
    }
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
#pragma unroll
        for (int p = 0; p < mfma_n_products<NSLICE>(); p++) {
          acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[MFMA_PROD_A[p]][fr], b_frag[MFMA_PROD_B[p]][fc],
              acc[fr][fc], 0, 0, 0);
        }
      }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }
}

}  // namespace raft_amd
