// CSR SpMV — wave64 sub-wave-per-row (BASELINE config 4's hot kernel).
//
// Reference parity (WHAT): raft/sparse cusparse SpMV usage (Lanczos inner
// loop, lanczos.cuh:305-314). MI355X design: HBM-bound at ~12 B/nnz; a
// sub-wave of SW lanes (power of two, picked host-side from mean nnz/row)
// covers one row so short rows (the 10-nnz/row BASELINE graph) don't idle
// 54/64 lanes, while long rows still get coalesced segment reads.

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

template <int SW, typename T>
__global__ void csr_spmv_kernel(const int* __restrict__ indptr,
                                const int* __restrict__ indices,
                                const T* __restrict__ values,
                                const T* __restrict__ x, T* __restrict__ y,
                                long long n_rows) {
  const long long rows_per_block = blockDim.x / SW;
  const int lane = threadIdx.x % SW;
  long long row = (long long)blockIdx.x * rows_per_block + threadIdx.x / SW;
  const long long stride = (long long)gridDim.x * rows_per_block;
  for (; row < n_rows; row += stride) {
    const int lo = indptr[row], hi = indptr[row + 1];
    T acc = T(0);
    for (int e = lo + lane; e < hi; e += SW) acc += values[e] * x[indices[e]];
    for (int off = SW >> 1; off > 0; off >>= 1) acc += __shfl_xor(acc, off, RAFT_AMD_WAVE);
    if (lane == 0) y[row] = acc;
  }
}

template <typename T>
void launch_csr_spmv(const int* indptr, const int* indices, const T* values,
                     const T* x, T* y, long long n_rows, long long nnz,
                     hipStream_t stream) {
  const long long mean = n_rows ? (nnz + n_rows - 1) / n_rows : 1;
  constexpr int BLOCK = 256;
  auto launch = [&](auto swc) {
    constexpr int SW = decltype(swc)::value;
    // NOT grid-capped: short rows make this latency-bound (indptr loads per
    // row); one sub-wave per row maximizes memory-level parallelism
    // (capped grid measured 746 GB/s effective @ 10 nnz/row)
    long long blocks = (n_rows * SW + BLOCK - 1) / BLOCK;
    if (blocks > 2147483647ll) blocks = 2147483647ll;
    hipLaunchKernelGGL((csr_spmv_kernel<SW, T>), dim3((unsigned)blocks), dim3(BLOCK),
                       0, stream, indptr, indices, values, x, y, n_rows);
  };
  if (mean <= 4) launch(std::integral_constant<int, 2>{});
  else if (mean <= 8) launch(std::integral_constant<int, 4>{});
  else if (mean <= 16) launch(std::integral_constant<int, 8>{});
  else if (mean <= 32) launch(std::integral_constant<int, 16>{});
  else if (mean <= 128) launch(std::integral_constant<int, 32>{});
  else launch(std::integral_constant<int, 64>{});
}

template void launch_csr_spmv<float>(const int*, const int*, const float*, const float*,
                                     float*, long long, long long, hipStream_t);
template void launch_csr_spmv<double>(const int*, const int*, const double*, const double*,
                                      double*, long long, long long, hipStream_t);

}  // namespace raft_amd

namespace raft_amd {

// SDDMM: vals[e] = dot(a[rows[e]], b[cols[e]]) — sampled dense-dense matmul
// (reference: cusparse SDDMM, sddmm.hpp:43). Sub-wave per edge (SW lanes
// cover d); no [nnz, d] gather temporaries (the composition path
// materializes 2*nnz*d floats — 100 GB at nnz=1e8, d=128).
template <int SW>
__global__ void sddmm_kernel(const float* __restrict__ a,
                             const float* __restrict__ b,
                             const int* __restrict__ rows,
                             const int* __restrict__ cols,
                             float* __restrict__ vals, long long nnz,
                             long long d) {
  const long long edges_per_block = blockDim.x / SW;
  const int lane = threadIdx.x % SW;
  long long e = (long long)blockIdx.x * edges_per_block + threadIdx.x / SW;
  const long long stride = (long long)gridDim.x * edges_per_block;
  for (; e < nnz; e += stride) {
    const float* ar = a + (long long)rows[e] * d;
    const float* br = b + (long long)cols[e] * d;
    float acc = 0.f;
    for (long long j = lane; j < d; j += SW) acc += ar[j] * br[j];
    for (int off = SW >> 1; off > 0; off >>= 1)
      acc += __shfl_xor(acc, off, RAFT_AMD_WAVE);
    if (lane == 0) vals[e] = acc;
  }
}

void launch_sddmm(const float* a, const float* b, const int* rows,
                  const int* cols, float* vals, long long nnz, long long d,
                  hipStream_t stream) {
  constexpr int BLOCK = 256;
  auto launch = [&](auto swc) {
    constexpr int SW = decltype(swc)::value;
    long long blocks = (nnz * SW + BLOCK - 1) / BLOCK;
    if (blocks > 1048576) blocks = 1048576;
    hipLaunchKernelGGL((sddmm_kernel<SW>), dim3((unsigned)blocks), dim3(BLOCK),
                       0, stream, a, b, rows, cols, vals, nnz, d);
  };
  if (d <= 8) launch(std::integral_constant<int, 4>{});
  else if (d <= 32) launch(std::integral_constant<int, 16>{});
  else if (d <= 128) launch(std::integral_constant<int, 32>{});
  else launch(std::integral_constant<int, 64>{});
}

}  // namespace raft_amd
