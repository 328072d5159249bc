// Fused L2 nearest-neighbor engines (csrc/fused_l2nn*.hip) — the k-means
// assignment step. dmin2 (second-best distance) feeds the verified engine's
// provable split-error margin test (see csrc/fused_l2nn.hip header).
#pragma once

#include "core.hpp"

namespace raft_amd {

// v1: one block per row tile, col-tile loop in-kernel
void launch_fused_l2nn_split(const void** xsl, const void** csl, const float* xn,
                             const float* cn, float* dmin, int* amin, float* dmin2,
                             long long m, int n, int d, int nslice, hipStream_t s);
// 2D XCD-swizzled tile-pair grid + partials combine (default at m>=1M):
// pd/pd2/pi are (n/128/GT * m) workspace arrays
bool fused_l2nn_2d_supported(int nslice, long long m, int n, int d);
void launch_fused_l2nn_2d(const void** xsl, const void** csl, const float* xn,
                          const float* cn, float* pd, float* pd2, int* pi,
                          float* dmin, int* amin, float* dmin2,
                          long long m, int n, int d, int nslice, hipStream_t s);
// 8-wave 128x256 variant (default for 3-slice) + persistent-X (env-gated)
bool fused_l2nn_w8_supported(int nslice, int n, int d);
void launch_fused_l2nn_w8(const void** xsl, const void** csl, const float* xn,
                          const float* cn, float* dmin, int* amin, float* dmin2,
                          long long m, int n, int d, int nslice, hipStream_t s);
// exact-fp32 rescan/repair of rows whose margin is inside the split bound;
// lead/tail select the bound for the split mode that produced dmin/dmin2
// (defaults = bf16x2v; bf16x1v passes 2^-7 / 2^-12)
void launch_l2nn_verify_repair(const float* x, const float* c, const float* xn,
                               float* dmin, int* amin, const float* dmin2,
                               const float* cn_max_dev, long long m, int n, int d,
                               hipStream_t s, float lead = 0x1p-13f,
                               float tail = 0x1p-18f);

}  // namespace raft_amd
