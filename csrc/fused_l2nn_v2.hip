// Fused L2-NN, persistent-X variant (the k-means assignment hot kernel).
// Reference parity: the same fusedL2NN contract as fused_l2nn.hip — these
// are alternative CDNA4 schedules (A/B results in BASELINE.md).
//
// v1 (fused_l2nn.hip) re-stages the X tile for every 128-centroid tile:
// X HBM traffic = n/128 full passes (measured ~60% of kernel time at
// n=1024, d=256). Here the workgroup's X tile lives in LDS for the WHOLE
// kernel (d <= 256): X is read from HBM exactly once, and only the tiny C
// tiles (centroids: L2-resident, ~1 MB total) are staged per step.
//
// Geometry: 512 threads = 8 waves in a WR x WC grid over a BM x 128 output
// tile; per-wave fragment grid RF x CF of 16x16 MFMAs. LDS: X = NSLICE *
// (d/64) * BM*64 bf16 tiles + C = NSLICE * [128][64] tiles (<= 160 KiB).
// Variants: NSLICE<=2 @ BM=128 (d<=256), NSLICE=3 @ BM=128 (d<=128) or
// BM=64 (d<=256).

#include <hip/hip_runtime.h>

#include "mfma_common.h"

namespace raft_amd {

// stage a [ROWS][64] bf16 tile with BLOCK threads (ROWS*128 bytes)
template <int ROWS, int BLOCK>
__device__ __forceinline__ void stage_rows(const __bf16* __restrict__ g,
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

template <int NSLICE, int BM, int WR, int WC>
__launch_bounds__(WR* WC * 64, 2)
__global__ void fused_l2nn_persist_kernel(
    const __bf16* __restrict__ x0, const __bf16* __restrict__ x1,
    const __bf16* __restrict__ x2, const __bf16* __restrict__ c0,
    const __bf16* __restrict__ c1, const __bf16* __restrict__ c2,
    const float* __restrict__ xn, const float* __restrict__ cn,
    float* __restrict__ dmin, int* __restrict__ amin, float* __restrict__ dmin2,
    long long m, int n, int d) {
  constexpr int BLOCK = WR * WC * 64;
  constexpr int RF = BM / (16 * WR);   // row fragments per wave
  constexpr int CF = 128 / (16 * WC);  // col fragments per wave
  extern __shared__ __bf16 smem[];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};

  const int k_tiles = d / 64;
  const int xtile_elems = BM * 64;
  __bf16* cs_base = smem + NSLICE * k_tiles * xtile_elems;

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w / WC, wc = w % WC;
  const long long row0 = (long long)blockIdx.x * BM;

  // ---- stage X once ------------------------------------------------------
  for (int s = 0; s < NSLICE; s++)
    for (int kt = 0; kt < k_tiles; kt++)
      stage_rows<BM, BLOCK>(xg[s], smem + (s * k_tiles + kt) * xtile_elems,
                            row0, (long long)kt * 64, d, m - 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  float best[RF][4];
  int bidx[RF][4];
#pragma unroll
  for (int a = 0; a < RF; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) { best[a][b] = INFINITY; bidx[a][b] = 0; }

  const int n_tiles = n / 128;
  for (int nt = 0; nt < n_tiles; nt++) {
    f32x4 acc[RF][CF];
#pragma unroll
    for (int a = 0; a < RF; a++)
#pragma unroll
      for (int b = 0; b < CF; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < k_tiles; kt++) {
#pragma unroll
      for (int s = 0; s < NSLICE; s++)
        stage_rows<128, BLOCK>(cg[s], cs_base + s * 8192,
                               (long long)nt * 128, (long long)kt * 64, d, n - 1);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();

#pragma unroll
      for (int kf = 0; kf < 2; kf++) {
        bf16x8 a_frag[NSLICE][RF], b_frag[NSLICE][CF];
        const int kbyte = (kf * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
        for (int fr = 0; fr < RF; fr++) {
          const int r = wr * (16 * RF) + fr * 16 + (lane & 15);
          const int byte = mfma_swz(r * 128 + kbyte);
#pragma unroll
          for (int s = 0; s < NSLICE; s++)
            a_frag[s][fr] = *reinterpret_cast<const bf16x8*>(
                (const char*)(smem + (s * k_tiles + kt) * xtile_elems) + byte);
        }
#pragma unroll
        for (int fc = 0; fc < CF; fc++) {
          const int c = wc * (16 * CF) + fc * 16 + (lane & 15);
          const int byte = mfma_swz(c * 128 + kbyte);
#pragma unroll
          for (int s = 0; s < NSLICE; s++)
            b_frag[s][fc] = *reinterpret_cast<const bf16x8*>(
                (const char*)(cs_base + s * 8192) + byte);
        }
#pragma unroll
        for (int fr = 0; fr < RF; fr++)
#pragma unroll
          for (int fc = 0; fc < CF; fc++) {
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

    // fold this tile's columns into the running per-row best
    const int col_base = nt * 128 + wc * (16 * CF);
#pragma unroll
    for (int fr = 0; fr < RF; fr++) {
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        float v = INFINITY;
        int vi = 0;
#pragma unroll
        for (int fc = 0; fc < CF; fc++) {
          const int col = col_base + fc * 16 + (lane & 15);
          const float s = cn[col] - 2.f * acc[fr][fc][reg];
          if (s < v) { v = s; vi = col; }
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) {
          const float ov = __shfl_xor(v, off, RAFT_AMD_WAVE);
          const int oi = __shfl_xor(vi, off, RAFT_AMD_WAVE);
          if (ov < v || (ov == v && oi < vi)) { v = ov; vi = oi; }
        }
        if (v < best[fr][reg] || (v == best[fr][reg] && vi < bidx[fr][reg])) {
          best[fr][reg] = v;
          bidx[fr][reg] = vi;
        }
      }
    }
  }

  // combine the WC column-slice waves per row through LDS, then write
  __syncthreads();
  float* comb_v = reinterpret_cast<float*>(smem);        // [WC][BM]
  int* comb_i = reinterpret_cast<int*>(comb_v + WC * BM);
  if ((lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < RF; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * (16 * RF) + fr * 16 + (lane >> 4) * 4 + reg;
        comb_v[wc * BM + rl] = best[fr][reg];
        comb_i[wc * BM + rl] = bidx[fr][reg];
      }
  }
  __syncthreads();
  if (wc == 0 && (lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < RF; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * (16 * RF) + fr * 16 + (lane >> 4) * 4 + reg;
        float v = comb_v[rl];
        int vi = comb_i[rl];
        for (int q = 1; q < WC; q++) {
          const float ov = comb_v[q * BM + rl];
          const int oi = comb_i[q * BM + rl];
          if (ov < v || (ov == v && oi < vi)) { v = ov; vi = oi; }
        }
        const long long row = row0 + rl;
        if (row < m) {
          dmin[row] = fmaxf(v + xn[row], 0.f);
          amin[row] = vi;
          if (dmin2) dmin2[row] = v + xn[row];  // best2 not tracked here: mark
        }
      }
  }
}

template <int NSLICE, int BM, int WR, int WC>
static void launch_persist(const __bf16* x0, const __bf16* x1, const __bf16* x2,
                           const __bf16* c0, const __bf16* c1, const __bf16* c2,
                           const float* xn, const float* cn, float* dmin, int* amin,
                           float* dmin2, long long m, int n, int d, hipStream_t stream) {
  const int grid = (int)((m + BM - 1) / BM);
  const size_t lds =
      (size_t)NSLICE * (d / 64) * BM * 64 * 2 + (size_t)NSLICE * 8192 * 2;
  static bool attr_set = false;
  if (!attr_set) {
    HIP_CHECK(hipFuncSetAttribute(
        (const void*)&fused_l2nn_persist_kernel<NSLICE, BM, WR, WC>,
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024));
    attr_set = true;
  }
  hipLaunchKernelGGL((fused_l2nn_persist_kernel<NSLICE, BM, WR, WC>), dim3(grid),
                     dim3(WR * WC * 64), lds, stream, x0, x1, x2, c0, c1, c2, xn,
                     cn, dmin, amin, dmin2, m, n, d);
}

// ---------------------------------------------------------------------------
// w8 variant: 512 threads (8 waves, 2x4), output tile 128 rows x 256 cols.
// Halves the X re-read count AND the barrier count vs the 4-wave 128x128
// kernel at the same 2-waves/SIMD occupancy (96 KiB LDS for NSLICE=2,
// 144 KiB for NSLICE=3 -> 1 block/CU of 8 waves). Tracks (best, second-best)
// like v1 so the verified engine works unchanged.
// ---------------------------------------------------------------------------

template <int NSLICE>
__launch_bounds__(512, 2)
__global__ void fused_l2nn_w8_kernel(
    const __bf16* __restrict__ x0, const __bf16* __restrict__ x1,
    const __bf16* __restrict__ x2, const __bf16* __restrict__ c0,
    const __bf16* __restrict__ c1, const __bf16* __restrict__ c2,
    const float* __restrict__ xn, const float* __restrict__ cn,
    float* __restrict__ dmin, int* __restrict__ amin, float* __restrict__ dmin2,
    long long m, int n, int d) {
  constexpr int BLOCK = 512;
  extern __shared__ __bf16 smem[];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 8192;                       // [128][64] per slice
    cs[s] = smem + NSLICE * 8192 + s * 16384;      // [256][64] per slice
  }

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 2, wc = w & 3;  // 2x4 wave grid
  const long long row0 = (long long)blockIdx.x * 128;

  float best[4][4], best2[4][4];
  int bidx[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) {
      best[a][b] = INFINITY; best2[a][b] = INFINITY; bidx[a][b] = 0;
    }

  const int n_tiles = n / 256;
  const int k_tiles = d / 64;
  for (int nt = 0; nt < n_tiles; nt++) {
    f32x4 acc[4][4];
#pragma unroll
    for (int a = 0; a < 4; a++)
#pragma unroll
      for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < k_tiles; kt++) {
#pragma unroll
      for (int s = 0; s < NSLICE; s++) {
        stage_rows<128, BLOCK>(xg[s], xs[s], row0, (long long)kt * 64, d, m - 1);
        stage_rows<256, BLOCK>(cg[s], cs[s], (long long)nt * 256,
                               (long long)kt * 64, d, n - 1);
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();

#pragma unroll
      for (int kf = 0; kf < 2; kf++) {
        bf16x8 a_frag[NSLICE][4], b_frag[NSLICE][4];
        const int kbyte = (kf * 32 + (lane >> 4) * 8) * 2;
#pragma unroll
        for (int fr = 0; fr < 4; fr++) {
          const int r = wr * 64 + fr * 16 + (lane & 15);
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

    const int col_base = nt * 256 + wc * 64;
#pragma unroll
    for (int fr = 0; fr < 4; fr++) {
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int col = col_base + fc * 16 + (lane & 15);
          const float s = cn[col] - 2.f * acc[fr][fc][reg];
          // second-best via one v_med3_f32 (invariant best <= best2)
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

  // ONE cross-lane top-2 reduce after the nt loop (lane column sets disjoint)
#pragma unroll
  for (int fr = 0; fr < 4; fr++) {
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
  }

  // combine the 4 column-slice waves per row through LDS, then write
  __syncthreads();
  float* comb_v = reinterpret_cast<float*>(smem);          // [4][128]
  int* comb_i = reinterpret_cast<int*>(comb_v + 512);      // [4][128]
  float* comb_v2 = reinterpret_cast<float*>(comb_i + 512); // [4][128]
  if ((lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;  // 0..127
        comb_v[wc * 128 + rl] = best[fr][reg];
        comb_i[wc * 128 + rl] = bidx[fr][reg];
        comb_v2[wc * 128 + rl] = best2[fr][reg];
      }
  }
  __syncthreads();
  if (wc == 0 && (lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;
        float v = comb_v[rl];
        int vi = comb_i[rl];
        float s2 = comb_v2[rl];
        for (int q = 1; q < 4; q++) {
          const float ov = comb_v[q * 128 + rl];
          const int oi = comb_i[q * 128 + rl];
          const float ov2 = comb_v2[q * 128 + rl];
          s2 = fminf(fminf(s2, ov2), fmaxf(v, ov));
          if (ov < v || (ov == v && oi < vi)) { v = ov; vi = oi; }
        }
        const long long row = row0 + rl;
        if (row < m) {
          dmin[row] = fmaxf(v + xn[row], 0.f);
          amin[row] = vi;
          if (dmin2) dmin2[row] = s2 + xn[row];
        }
      }
  }
}

template <int NSLICE>
static void launch_w8(const __bf16* x0, const __bf16* x1, const __bf16* x2,
                      const __bf16* c0, const __bf16* c1, const __bf16* c2,
                      const float* xn, const float* cn, float* dmin, int* amin,
                      float* dmin2, long long m, int n, int d, hipStream_t stream) {
  const int grid = (int)((m + 127) / 128);
  const size_t lds = (size_t)NSLICE * (8192 + 16384) * sizeof(__bf16);
  static bool attr_set = false;
  if (!attr_set) {
    HIP_CHECK(hipFuncSetAttribute((const void*)&fused_l2nn_w8_kernel<NSLICE>,
                                  hipFuncAttributeMaxDynamicSharedMemorySize,
                                  160 * 1024));
    attr_set = true;
  }
  hipLaunchKernelGGL((fused_l2nn_w8_kernel<NSLICE>), dim3(grid), dim3(512), lds,
                     stream, x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2,
                     m, n, d);
}

bool fused_l2nn_w8_supported(int nslice, int n, int d) {
  // measured (10M x 256 k=1024): w8 wins only for NSLICE=3 (staging-bound:
  // halved X re-reads beat the lost cross-block barrier overlap); the 4-wave
  // 2-block v1 kernel wins for NSLICE<=2 (34.3 vs 38.0 ms/step).
  // RAFT_AMD_L2NN_W8=2 forces w8 for ALL nslice (A/B experiments).
  static const bool force_all = [] {
    const char* e = getenv("RAFT_AMD_L2NN_W8");
    return e && e[0] == '2';
  }();
  if (n % 256 != 0 || d % 64 != 0) return false;
  return force_all || nslice == 3;
}

void launch_fused_l2nn_w8(const void** xsl, const void** csl, const float* xn,
                          const float* cn, float* dmin, int* amin, float* dmin2,
                          long long m, int n, int d, int nslice, hipStream_t stream) {
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  if (nslice == 1)
    launch_w8<1>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
  else if (nslice == 2)
    launch_w8<2>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
  else
    launch_w8<3>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
}

// returns true when a persistent variant covers (nslice, d)
bool fused_l2nn_persist_supported(int nslice, int d) {
  if (d % 64 != 0) return false;
  if (nslice <= 2) return d <= 256;
  return d <= 256;  // NSLICE=3 uses BM=64 for d in (128, 256]
}

void launch_fused_l2nn_persist(const void** xsl, const void** csl, const float* xn,
                               const float* cn, float* dmin, int* amin, float* dmin2,
                               long long m, int n, int d, int nslice,
                               hipStream_t stream) {
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  if (nslice == 1) {
    launch_persist<1, 128, 2, 4>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
  } else if (nslice == 2) {
    launch_persist<2, 128, 2, 4>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
  } else if (d <= 128) {
    launch_persist<3, 128, 2, 4>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
  } else {
    launch_persist<3, 64, 1, 8>(x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d, stream);
  }
}

}  // namespace raft_amd
