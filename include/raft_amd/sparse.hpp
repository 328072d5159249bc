// Sparse kernels (csrc/spmv.hip): CSR SpMV with sub-wave-per-row mapping
// (2..64 lanes chosen by mean nnz/row), uncapped grid.
#pragma once

#include "core.hpp"

namespace raft_amd {

template <typename T>
void launch_csr_spmv(const int* indptr, const int* indices, const T* values, const T* x,
                     T* y, long long n_rows, long long nnz, hipStream_t s);

// sampled dense-dense matmul: vals[e] = dot(a[rows[e]], b[cols[e]]),
// a [m, d] / b [n, d] row-major fp32 (csrc/spmv.hip)
void launch_sddmm(const float* a, const float* b, const int* rows,
                  const int* cols, float* vals, long long nnz, long long d,
                  hipStream_t stream);

}  // namespace raft_amd
