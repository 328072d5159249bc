// Sparse kernels (csrc/spmv.hip): CSR SpMV with sub-wave-per-row mapping
// (2..64 lanes chosen by mean nnz/row), uncapped grid.
#pragma once

#include "core.hpp"

namespace raft_amd {

template <typename T>
void launch_csr_spmv(const int* indptr, const int* indices, const T* values, const T* x,
                     T* y, long long n_rows, long long nnz, hipStream_t s);

}  // namespace raft_amd
