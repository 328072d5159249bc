// Fused L2-NN, 256x256-tile engine over the BK=32 product-phase
// counted-vmcnt K-loop (mfma_common.h mfma256_bk32_kloop; design history and
// measured A/B in profiles/pmc_l2nn_256_ab.txt — on THIS op the round-1
// 128^2 2-blocks/CU engine stays faster, so this engine ships disabled; it
// is the production K-loop for the pairwise tile kernel, whose old schedule
// was 1 block/CU WITH full drains).
//
// Epilogue: fused top-2 argmin (never materializes the distance tile):
// per-lane top-2 over the tile columns, 16-lane shuffle merge, [256][17]
// LDS wave-combine, per-(row, col-tile) partials merged by
// l2nn_combine_partials_kernel (fused_l2nn_2d.hip).
//
// Reference parity: RAFT's fusedL2NN (k-means assignment), the contraction
// engine role of linalg/detail/contractions.cuh:140-307.

#include <hip/hip_runtime.h>

#include "mfma_common.h"

namespace raft_amd {

template <int NSLICE>
__launch_bounds__(512, 1)
__global__ void fused_l2nn_256_kernel(const __bf16* __restrict__ x0,
                                      const __bf16* __restrict__ x1,
                                      const __bf16* __restrict__ c0,
                                      const __bf16* __restrict__ c1,
                                      const float* __restrict__ cn,
                                      float* __restrict__ pd,
                                      float* __restrict__ pd2,
                                      int* __restrict__ pi,
                                      long long m, int n, int d, int n_groups) {
  extern __shared__ __bf16 smem[];

  // XCD-contiguous bijective remap (see fused_l2nn_2d.hip)
  const int nwg = gridDim.x;
  const int bid = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = bid & 7, slot = bid >> 3;
  const int t = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  const long long row0 = (long long)(t / n_groups) * 256;
  const int grp = t % n_groups;
  const long long col0 = (long long)grp * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wm = w >> 2, wn = w & 3;   // 2 x 4 wave grid

  Mfma256BK32 st;
  mfma256_bk32_setup(st, row0, col0, d, m - 1, n - 1, wm, wn, lane);

  f32x4 acc[8][4];
#pragma unroll
  for (int a = 0; a < 8; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) acc[a][b] = f32x4{0.f, 0.f, 0.f, 0.f};

  mfma256_bk32_kloop<NSLICE>(x0, x1, c0, c1, smem, st, acc, d / 32);

  // ---- fused top-2 argmin epilogue ----------------------------------------
  // per-lane top-2 over this thread's 4 columns per (fr, reg) row slot
  float best[8][4], best2[8][4];
  int bidx[8][4];
#pragma unroll
  for (int fr = 0; fr < 8; fr++)
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      float v = INFINITY, v2 = INFINITY;
      int vi = 0;
#pragma unroll
      for (int fc = 0; fc < 4; fc++) {
        const int col = (int)col0 + wn * 64 + fc * 16 + (lane & 15);
        const float s = cn[col] - 2.f * acc[fr][fc][reg];
        v2 = __builtin_amdgcn_fmed3f(s, v, v2);  // second-best in one op
        if (s < v) {
          v = s;
          vi = col;
        }
      }
      best[fr][reg] = v;
      best2[fr][reg] = v2;
      bidx[fr][reg] = vi;
    }

  // 16-lane shuffle merge (columns disjoint across lanes in a group)
#pragma unroll
  for (int fr = 0; fr < 8; fr++)
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

  // wave-combine through LDS: [256 rows][17] per value (wn = 4 candidates)
  __syncthreads();  // K-loop LDS is dead; reuse
  float* lv = reinterpret_cast<float*>(smem);    // [256][17]
  float* lv2 = lv + 256 * 17;
  int* li = reinterpret_cast<int*>(lv2 + 256 * 17);
  if ((lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < 8; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wm * 128 + fr * 16 + ((lane >> 4) & 3) * 4 + reg;
        lv[rl * 17 + wn] = best[fr][reg];
        lv2[rl * 17 + wn] = best2[fr][reg];
        li[rl * 17 + wn] = bidx[fr][reg];
      }
  }
  __syncthreads();
  const int rl = tid;
  if (rl < 256) {
    const long long row = row0 + rl;
    if (row < m) {
      float v = INFINITY, v2 = INFINITY;
      int vi = 0;
#pragma unroll
      for (int c = 0; c < 4; c++) {
        const float b = lv[rl * 17 + c];
        const float b2 = lv2[rl * 17 + c];
        const int bi = li[rl * 17 + c];
        v2 = fminf(fminf(v2, b2), fmaxf(v, b));
        if (b < v || (b == v && bi < vi)) { v = b; vi = bi; }
      }
      const long long o = (long long)grp * m + row;
      pd[o] = v;
      pd2[o] = v2;
      pi[o] = vi;
    }
  }
}

// combine launch shared with the 2D engine (defined in fused_l2nn_2d.hip)
void launch_l2nn_combine(const float* pd, const float* pd2, const int* pi,
                         const float* xn, float* dmin, int* amin, float* dmin2,
                         long long m, int n_groups, hipStream_t stream);

bool fused_l2nn_256_supported(int nslice, long long m, int n, int d) {
  static const char mode = [] {
    const char* e = getenv("RAFT_AMD_L2NN_256");
    return e ? e[0] : 'a';
  }();
  if (mode == '0') return false;
  if (nslice > 2) return false;
  if (n % 256 != 0 || d % 32 != 0) return false;
  if (mode == '1') return true;
  // auto: OFF. Measured (round 2, profiles/pmc_l2nn_256_ab.txt): the 256^2
  // counted-vmcnt schedule at 1 block/CU loses to the 128^2 2-blocks/CU
  // implicit overlap on this op at every shape tried (23.7 vs 16.8 ms at
  // 10M x 1024 d=256 after fixing LDS conflicts to 0, removing K-loop
  // spills, and hoisting addresses). Kept (exact results, agree=1.0) as the
  // base for future schedule work; enable with RAFT_AMD_L2NN_256=1.
  return false;
}

void launch_fused_l2nn_256(const void** xsl, const void** csl, const float* xn,
                           const float* cn, float* pd, float* pd2, int* pi,
                           float* dmin, int* amin, float* dmin2, long long m,
                           int n, int d, int nslice, hipStream_t stream) {
  const int n_row_tiles = (int)((m + 255) / 256);
  const int n_groups = n / 256;
  const int grid = n_row_tiles * n_groups;
  const size_t lds_kloop = 8 * 16384;                 // 128 KiB
  const size_t lds_epi = 3 * 256 * 17 * 4;            // 52 KiB
  const size_t lds = lds_kloop > lds_epi ? lds_kloop : lds_epi;
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  if (nslice == 1) {
    hipLaunchKernelGGL((fused_l2nn_256_kernel<1>), dim3(grid), dim3(512), lds,
                       stream, x0, x1, c0, c1, cn, pd, pd2, pi, m, n, d,
                       n_groups);
  } else {
    hipLaunchKernelGGL((fused_l2nn_256_kernel<2>), dim3(grid), dim3(512), lds,
                       stream, x0, x1, c0, c1, cn, pd, pd2, pi, m, n, d,
                       n_groups);
  }
  launch_l2nn_combine(pd, pd2, pi, xn, dmin, amin, dmin2, m, n_groups, stream);
}

}  // namespace raft_amd
