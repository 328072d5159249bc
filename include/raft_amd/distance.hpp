// Pairwise distances (csrc/pairwise.hip, csrc/pairwise_mfma.hip).
// The fp32 L2 path runs as split-bf16 slice products on MFMA (no fp32 MFMA
// on CDNA4): xsl/csl are nslice in {1,2,3} bf16 slice pointers.
#pragma once

#include "core.hpp"

namespace raft_amd {

void launch_l2_epilogue(float* g, const float* xn, const float* yn, long long m,
                        long long n, hipStream_t s);
void launch_l2nn_epilogue(const float* g, const float* xn, const float* yn,
                          float* dmin, int* amin, long long m, long long n, hipStream_t s);
// code = DistanceCode
void launch_pairwise_unexpanded(const float* x, const float* y, float* out, long long m,
                                long long n, long long d, int code, float p, hipStream_t s);
// 128x128-tile MFMA kernel; out[ldo] row stride; epilogue fused into the
// single non-temporal C write
void launch_pairwise_l2_mfma(const void** xsl, const void** csl, const float* xn,
                             const float* yn, float* out, long long m, long long n,
                             int d, long long ldo, int nslice, bool sqrt_out, hipStream_t s);
void launch_pairwise_l2_mfma256(const void** xsl, const void** csl, const float* xn,
                                const float* yn, float* out, long long m, long long n,
                                int d, long long ldo, int nslice, bool sqrt_out,
                                hipStream_t s);
// filtered variant (kNN candidate emission below per-row thresholds)
void launch_pairwise_l2_filter(const void** xsl, const void** csl, const float* xn,
                               const float* yn, const float* thresh, float* out_d,
                               int* out_i, int* counts, int cap, long long col_offset,
                               long long m, long long n, int d, int nslice,
                               hipStream_t s);

}  // namespace raft_amd
