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
__device__ __forceinline__ void xcd_supertile_decode(int rg, long long* rt,
                                                     long long* ct) {
  const int nwg = gridDim.x;
  const int bid = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, slot = bid >> 3;
  const int t = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  const int g = t >> 6, w = t & 63;
  *rt = (long long)(g % rg) * 8 + (w & 7);
  *ct = (long long)(g / rg) * 8 + (w >> 3);
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
