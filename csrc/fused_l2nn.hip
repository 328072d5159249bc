// Fused L2 nearest-neighbor: split-bf16 MFMA distance + in-register argmin.
//
// Reference parity (WHAT): RAFT's historical fused-L2-NN (contraction engine
// + key-value argmin epilogue) — the k-means assignment step and BASELINE
// config 5. The m x n distance matrix NEVER touches HBM.
//
// MI355X design (why this shape — see cdna_hip_programming.md §5):
//  * CDNA4 has no fp32 MFMA. fp32 x is pre-split into NSLICE bf16 slices
//    (x = x0 + x1 [+ x2]); the dot term is the sum of slice-product MFMAs
//    accumulated into ONE fp32 AGPR tile:
//      NSLICE=1: bf16 input (3 per-element bits) — the bf16 data path
//      NSLICE=2: 3 products, ~2^-16 accuracy (TF32-class)
//      NSLICE=3: 6 products, fp32-class (~2^-24)
//    at bf16 MFMA rate (2.5 PF dense) instead of the 157 TF fp32 vector ALU.
//  * Tile: 128 rows x 128 centroids per 256-thread block (4 waves, 2x2),
//    K-step 64. LDS = NSLICE * 32 KiB (dynamic).
//  * Staging via __builtin_amdgcn_global_load_lds width=16 (direct HBM->LDS).
//    LDS rows are 128 B — a 32-way bank conflict for ds_read_b128 column
//    slices — so tiles are XOR-swizzled (byte ^= (row&7)<<4): gload_lds
//    writes linearly, therefore the SOURCE address is inverse-swizzled
//    per-lane and reads apply the same XOR (both-sides-or-neither rule).
//  * argmin epilogue: MFMA C/D layout (col=lane&15, row=(lane>>4)*4+reg) puts
//    each output row across the 16 lanes of a row-group: 4 xor-shuffles
//    reduce (score, index) pairs; the running best stays in registers across
//    centroid tiles. score = ||c||^2 - 2 x.c (||x||^2 added only at the
//    final write — it does not affect the argmin).

#include <hip/hip_runtime.h>

#include "mfma_common.h"

namespace raft_amd {

template <int NSLICE, bool DB = false, bool PHASED = false>
__launch_bounds__(256, 2)
__global__ void fused_l2nn_kernel(const __bf16* __restrict__ x0,
                                  const __bf16* __restrict__ x1,
                                  const __bf16* __restrict__ x2,
                                  const __bf16* __restrict__ c0,
                                  const __bf16* __restrict__ c1,
                                  const __bf16* __restrict__ c2,
                                  const float* __restrict__ xn,
                                  const float* __restrict__ cn,
                                  float* __restrict__ dmin, int* __restrict__ amin,
                                  float* __restrict__ dmin2,
                                  long long m, int n, int d) {
  extern __shared__ __bf16 smem[];
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
  __bf16* xs2[2][NSLICE];
  __bf16* cs2[2][NSLICE];
  const __bf16* const xg[3] = {x0, x1, x2};
  const __bf16* const cg[3] = {c0, c1, c2};
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 8192;
    cs[s] = smem + (NSLICE + s) * 8192;
#pragma unroll
    for (int b = 0; b < 2; b++) {
      xs2[b][s] = smem + (b * 2 * NSLICE + s) * 4096;
      cs2[b][s] = smem + (b * 2 * NSLICE + NSLICE + s) * 4096;
    }
  }

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;  // 2x2 wave grid
  const long long row0 = (long long)blockIdx.x * 128;

  float best[4][4], best2[4][4];
  int bidx[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) { best[a][b] = INFINITY; best2[a][b] = INFINITY; bidx[a][b] = 0; }

  const int n_tiles = n / 128;

  for (int nt = 0; nt < n_tiles; nt++) {
    f32x4 acc[4][4];
#pragma unroll
    for (int a = 0; a < 4; a++)
#pragma unroll
      for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

    if constexpr (PHASED && NSLICE == 2) {
      mfma_tile_kloop_p2(xg, cg, xs, cs, acc, row0, (long long)nt * 128, d,
                         m - 1, n - 1, wr, wc, lane);
    } else if constexpr (DB) {
      mfma_tile_kloop_db32<NSLICE>(xg, cg, xs2, cs2, acc, row0,
                                   (long long)nt * 128, d, m - 1, n - 1, wr, wc,
                                   lane);
    } else {
      mfma_tile_kloop<NSLICE>(xg, cg, xs, cs, acc, row0, (long long)nt * 128, d,
                              m - 1, n - 1, wr, wc, lane);
    }

    // epilogue: fold this tile's columns into each lane's LOCAL running
    // (best, second-best) — lanes own disjoint column sets, so no cross-lane
    // work is needed per tile; ONE shuffle reduce happens after the nt loop
    // (the per-tile 12-shuffle reduce was ~1/3 of the kernel's non-MFMA time).
    const int col_base = nt * 128 + wc * 64;
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

  // ONE cross-lane top-2 reduce per (fr, reg) over the 16 lanes of the
  // row-group (index-aware merge: shared-best lanes cannot donate their
  // loser as a second-best candidate)
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

  // combine the two column-half waves (wc = 0,1) of each row through LDS,
  // then one wave per row-half writes. smem is free after the last barrier.
  __syncthreads();
  float* comb_v = reinterpret_cast<float*>(smem);          // [2][128]
  int* comb_i = reinterpret_cast<int*>(comb_v + 256);      // [2][128]
  float* comb_v2 = reinterpret_cast<float*>(comb_i + 256); // [2][128]
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
        float v0 = comb_v[rl], v1 = comb_v[128 + rl];
        int i0 = comb_i[rl], i1 = comb_i[128 + rl];
        const float s2 = fminf(fminf(comb_v2[rl], comb_v2[128 + rl]),
                               fmaxf(v0, v1));
        const bool take1 = (v1 < v0) || (v1 == v0 && i1 < i0);
        const float v = take1 ? v1 : v0;
        const int vi = take1 ? i1 : i0;
        const long long row = row0 + rl;
        if (row < m) {
          dmin[row] = fmaxf(v + xn[row], 0.f);
          amin[row] = vi;
          if (dmin2) dmin2[row] = s2 + xn[row];  // second-best, unclamped
        }
      }
  }
}

bool l2nn_phased() {
  static const bool on = [] {
    const char* e = getenv("RAFT_AMD_L2NN_PHASED");
    return e && e[0] == '1';
  }();
  return on;
}

void launch_fused_l2nn_split(const void** xsl, const void** csl, const float* xn,
                             const float* cn, float* dmin, int* amin, float* dmin2,
                             long long m, int n, int d, int nslice,
                             hipStream_t stream) {
  const int grid = (int)((m + 127) / 128);
  const size_t lds = (size_t)nslice * 2 * 8192 * sizeof(__bf16);
  static const bool use_db = [] {
    const char* e = getenv("RAFT_AMD_L2NN_DB");
    return e && e[0] == '1';
  }();
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  switch (nslice) {
    case 1:
      if (use_db)
        hipLaunchKernelGGL((fused_l2nn_kernel<1, true>), dim3(grid), dim3(256), lds,
                           stream, x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      else
        hipLaunchKernelGGL((fused_l2nn_kernel<1>), dim3(grid), dim3(256), lds, stream,
                           x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      break;
    case 2:
      if (use_db)
        hipLaunchKernelGGL((fused_l2nn_kernel<2, true>), dim3(grid), dim3(256), lds,
                           stream, x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      else if (l2nn_phased())
        hipLaunchKernelGGL((fused_l2nn_kernel<2, false, true>), dim3(grid), dim3(256),
                           lds, stream, x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      else
        hipLaunchKernelGGL((fused_l2nn_kernel<2>), dim3(grid), dim3(256), lds, stream,
                           x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      break;
    case 3: {
      static bool attr_set = false;
      if (!attr_set) {
        HIP_CHECK(hipFuncSetAttribute((const void*)&fused_l2nn_kernel<3>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      96 * 1024));
        HIP_CHECK(hipFuncSetAttribute((const void*)&fused_l2nn_kernel<3, true>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      96 * 1024));
        attr_set = true;
      }
      if (use_db)
        hipLaunchKernelGGL((fused_l2nn_kernel<3, true>), dim3(grid), dim3(256), lds,
                           stream, x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      else
        hipLaunchKernelGGL((fused_l2nn_kernel<3>), dim3(grid), dim3(256), lds, stream,
                           x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, dmin2, m, n, d);
      break;
    }
    default:
      throw std::runtime_error("fused_l2nn: nslice must be 1, 2 or 3");
  }
}

}  // namespace raft_amd
