// Row/column reductions, argmin, normalization (csrc/reductions.hip).
// CDNA4 design: logical-warp widths {2..64} of wave64, Kahan-compensated
// sums, float4 row loads; column reduce is a 2D (col-tile x row-tile) grid
// with atomic combine and a sub-wave mapping for skinny d.
#pragma once

#include "core.hpp"

namespace raft_amd {

// op template parameter = ReduceOpCode value
template <int OP, typename T>
void launch_reduce_rows(const T* x, T* out, long long n_rows, long long d, hipStream_t s);
template <int OP, typename T>
void launch_reduce_cols(const T* x, T* out, long long n_rows, long long d, hipStream_t s);
void launch_row_argmin(const float* x, int* out, long long n_rows, long long d, hipStream_t s);
void launch_row_normalize_l2(const float* x, float* out, long long n_rows, long long d,
                             float eps, hipStream_t s);
// squared L2 row norms of a bf16 matrix accumulated in fp32 (no fp32 copy)
void launch_rows_sqnorm_bf16(const void* x, float* out, long long n_rows, long long d,
                             hipStream_t s);

}  // namespace raft_amd
