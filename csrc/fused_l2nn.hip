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

#include "common.h"

namespace raft_amd {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GLOAD_LDS(gp, lp)                                                      \
  __builtin_amdgcn_global_load_lds(                                           \
      (const __attribute__((address_space(1))) void*)(gp),                    \
      (__attribute__((address_space(3))) void*)(lp), 16, 0, 0)

// swizzle: flip byte-offset bit4 by row bits (rows are 128 B = 64 bf16)
__device__ __forceinline__ int swz(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

// stage a [128][64] bf16 tile from row-major global (leading dim ld elements)
// into a swizzled LDS tile (16 KiB). 256 threads, 4 gload rounds.
__device__ __forceinline__ void stage_tile128(const __bf16* __restrict__ g,
                                              __bf16* lds, long long row0,
                                              long long k0, long long ld,
                                              long long max_row) {
  const int t = threadIdx.x;
  const int w = t / RAFT_AMD_WAVE;
#pragma unroll
  for (int j = 0; j < 4; j++) {
    const int o = j * 4096 + t * 16;  // linear dest byte
    const int o_src = swz(o);         // fetch what belongs here
    long long r = row0 + (o_src >> 7);
    if (r > max_row) r = max_row;
    const long long goff = r * ld + k0 + ((o_src & 127) >> 1);
    // LDS dest = wave-uniform base + lane*16 (HW rule); global source is
    // per-lane. Our dest o = (j*4096 + w*1024) + lane*16 by construction.
    __bf16* lbase = lds + (j * 4096 + w * 1024) / 2;
    GLOAD_LDS(g + goff, lbase);
  }
}

// slice-product list per NSLICE (see header comment)
template <int NSLICE>
__device__ __forceinline__ constexpr int n_products() {
  return NSLICE == 1 ? 1 : (NSLICE == 2 ? 3 : 6);
}

__device__ constexpr int PROD_A[6] = {0, 0, 1, 1, 0, 2};
__device__ constexpr int PROD_B[6] = {0, 1, 0, 1, 2, 0};

template <int NSLICE>
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
                                  long long m, int n, int d) {
  extern __shared__ __bf16 smem[];
  __bf16* xs[NSLICE];
  __bf16* cs[NSLICE];
  const __bf16* xg[3] = {x0, x1, x2};
  const __bf16* cg[3] = {c0, c1, c2};
#pragma unroll
  for (int s = 0; s < NSLICE; s++) {
    xs[s] = smem + s * 8192;
    cs[s] = smem + (NSLICE + s) * 8192;
  }

  const int lane = threadIdx.x % RAFT_AMD_WAVE;
  const int w = threadIdx.x / RAFT_AMD_WAVE;
  const int wr = w >> 1, wc = w & 1;  // 2x2 wave grid
  const long long row0 = (long long)blockIdx.x * 128;

  float best[4][4];
  int bidx[4][4];
#pragma unroll
  for (int a = 0; a < 4; a++)
#pragma unroll
    for (int b = 0; b < 4; b++) { best[a][b] = INFINITY; bidx[a][b] = 0; }

  const int n_tiles = n / 128;
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
        stage_tile128(xg[s], xs[s], row0, (long long)kt * 64, d, m - 1);
        stage_tile128(cg[s], cs[s], (long long)nt * 128, (long long)kt * 64, d, n - 1);
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();

#pragma unroll
      for (int kf = 0; kf < 2; kf++) {
        bf16x8 a_frag[NSLICE][4], b_frag[NSLICE][4];
#pragma unroll
        for (int fr = 0; fr < 4; fr++) {
          const int r = wr * 64 + fr * 16 + (lane & 15);
          const int byte = swz(r * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
#pragma unroll
          for (int s = 0; s < NSLICE; s++)
            a_frag[s][fr] = *reinterpret_cast<const bf16x8*>((const char*)xs[s] + byte);
        }
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
          const int c = wc * 64 + fc * 16 + (lane & 15);
          const int byte = swz(c * 128 + (kf * 32 + (lane >> 4) * 8) * 2);
#pragma unroll
          for (int s = 0; s < NSLICE; s++)
            b_frag[s][fc] = *reinterpret_cast<const bf16x8*>((const char*)cs[s] + byte);
        }
#pragma unroll
        for (int fr = 0; fr < 4; fr++)
#pragma unroll
          for (int fc = 0; fc < 4; fc++) {
#pragma unroll
            for (int p = 0; p < n_products<NSLICE>(); p++) {
              acc[fr][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  a_frag[PROD_A[p]][fr], b_frag[PROD_B[p]][fc], acc[fr][fc],
                  0, 0, 0);
            }
          }
      }
      __syncthreads();
    }

    // epilogue: fold this tile's 64 columns-per-wave into the running best.
    const int col_base = nt * 128 + wc * 64;
#pragma unroll
    for (int fr = 0; fr < 4; fr++) {
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        float v = INFINITY;
        int vi = 0;
#pragma unroll
        for (int fc = 0; fc < 4; fc++) {
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

  // combine the two column-half waves (wc = 0,1) of each row through LDS,
  // then one wave per row-half writes. smem is free after the last barrier.
  __syncthreads();
  float* comb_v = reinterpret_cast<float*>(smem);          // [2][128]
  int* comb_i = reinterpret_cast<int*>(comb_v + 256);      // [2][128]
  if ((lane & 15) == 0) {
#pragma unroll
    for (int fr = 0; fr < 4; fr++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int rl = wr * 64 + fr * 16 + (lane >> 4) * 4 + reg;  // 0..127
        comb_v[wc * 128 + rl] = best[fr][reg];
        comb_i[wc * 128 + rl] = bidx[fr][reg];
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
        const bool take1 = (v1 < v0) || (v1 == v0 && i1 < i0);
        const float v = take1 ? v1 : v0;
        const int vi = take1 ? i1 : i0;
        const long long row = row0 + rl;
        if (row < m) {
          dmin[row] = fmaxf(v + xn[row], 0.f);
          amin[row] = vi;
        }
      }
  }
}

void launch_fused_l2nn_split(const void** xsl, const void** csl, const float* xn,
                             const float* cn, float* dmin, int* amin, long long m,
                             int n, int d, int nslice, hipStream_t stream) {
  const int grid = (int)((m + 127) / 128);
  const size_t lds = (size_t)nslice * 2 * 8192 * sizeof(__bf16);
  const __bf16* x0 = (const __bf16*)xsl[0];
  const __bf16* x1 = (const __bf16*)(nslice > 1 ? xsl[1] : xsl[0]);
  const __bf16* x2 = (const __bf16*)(nslice > 2 ? xsl[2] : xsl[0]);
  const __bf16* c0 = (const __bf16*)csl[0];
  const __bf16* c1 = (const __bf16*)(nslice > 1 ? csl[1] : csl[0]);
  const __bf16* c2 = (const __bf16*)(nslice > 2 ? csl[2] : csl[0]);
  switch (nslice) {
    case 1:
      hipLaunchKernelGGL((fused_l2nn_kernel<1>), dim3(grid), dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, m, n, d);
      break;
    case 2:
      hipLaunchKernelGGL((fused_l2nn_kernel<2>), dim3(grid), dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, m, n, d);
      break;
    case 3: {
      static bool attr_set = false;
      if (!attr_set) {
        HIP_CHECK(hipFuncSetAttribute((const void*)&fused_l2nn_kernel<3>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      96 * 1024));
        attr_set = true;
      }
      hipLaunchKernelGGL((fused_l2nn_kernel<3>), dim3(grid), dim3(256), lds, stream,
                         x0, x1, x2, c0, c1, c2, xn, cn, dmin, amin, m, n, d);
      break;
    }
    default:
      throw std::runtime_error("fused_l2nn: nslice must be 1, 2 or 3");
  }
}

}  // namespace raft_amd
