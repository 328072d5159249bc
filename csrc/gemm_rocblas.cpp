// rocBLAS gemm_ex wrapper: bf16 x bf16 -> fp32 with fp32 accumulate.
//
// Reference parity: raft/linalg/detail/cublas_wrappers.hpp cublasgemmEx usage.
// This is the building block of the split-bf16 fp32 GEMM emulation
// (raft_amd/linalg/gemm.py): torch.matmul on bf16 rounds the output to bf16,
// which destroys the splitting scheme — rocblas_gemm_ex keeps C in fp32.
// The MFMA path (2.5 PF dense bf16) is what rocBLAS dispatches to on gfx950.

#include <rocblas/rocblas.h>

#include <algorithm>
#include <mutex>
#include <stdexcept>
#include <string>

#define ROCBLAS_CHECK(expr)                                                   \
  do {                                                                        \
    rocblas_status _s = (expr);                                               \
    if (_s != rocblas_status_success) {                                       \
      throw std::runtime_error(std::string("rocBLAS error ") +                \
                               std::to_string((int)_s) + " at " __FILE__ ":" + \
                               std::to_string(__LINE__));                     \
    }                                                                         \
  } while (0)

namespace raft_amd {

static rocblas_handle get_handle() {
  static rocblas_handle h = nullptr;
  static std::once_flag flag;
  std::call_once(flag, [] { ROCBLAS_CHECK(rocblas_create_handle(&h)); });
  return h;
}

// rocblas_gemm_ex corrupts outputs with >= 2^31 elements on this stack
// (measured 2026-09: a [256, 30M] fp32 C has exact rows 0..~200 and garbage
// near the tail — a 32-bit C-element index in the vendor epilogue;
// rocblas_sgemm is NOT affected). Every gemm_ex wrapper row-chunks below
// the boundary; our own HIP kernels index 64-bit and never hit this.
static constexpr long long kMaxGemmExElems = 1ll << 30;

// C[m,n] (row-major, fp32) = A[m,k] (row-major bf16) @ B[k,n] (row-major bf16)
//                            + beta * C
// Row-major is expressed as the transposed column-major problem:
// C^T = B^T A^T with everything column-major.
void gemm_bf16_f32_rowmajor(const void* a, const void* b, float* c, long long m,
                            long long n, long long k, float beta, void* stream) {
  if (m * n > kMaxGemmExElems && m > 1) {
    const long long rows = std::max(1ll, kMaxGemmExElems / n);
    for (long long r0 = 0; r0 < m; r0 += rows)
      gemm_bf16_f32_rowmajor(
          static_cast<const unsigned short*>(a) + r0 * k, b, c + r0 * n,
          std::min(rows, m - r0), n, k, beta, stream);
    return;
  }
  rocblas_handle h = get_handle();
  ROCBLAS_CHECK(rocblas_set_stream(h, (hipStream_t)stream));
  const float alpha = 1.0f;
  ROCBLAS_CHECK(rocblas_gemm_ex(
      h, rocblas_operation_none, rocblas_operation_none,
      (rocblas_int)n, (rocblas_int)m, (rocblas_int)k, &alpha,
      b, rocblas_datatype_bf16_r, (rocblas_int)n,
      a, rocblas_datatype_bf16_r, (rocblas_int)k, &beta,
      c, rocblas_datatype_f32_r, (rocblas_int)n,
      c, rocblas_datatype_f32_r, (rocblas_int)n,
      rocblas_datatype_f32_r, rocblas_gemm_algo_standard, 0, 0));
}

// C[m,n] (row-major fp32) = A[m,k] @ B[n,k]^T + beta*C  (both row-major) —
// the natural pairwise-distance form: no transposed copies.
// Column-major: C_cm[n,m] = B_cm^T[n,k] * A_cm[k,m].
void gemm_bf16_f32_nt_rowmajor(const void* a, const void* b, float* c, long long m,
                               long long n, long long k, float beta, void* stream) {
  if (m * n > kMaxGemmExElems && m > 1) {
    // see kMaxGemmExElems: vendor gemm_ex 32-bit C-index overflow guard
    const long long rows = std::max(1ll, kMaxGemmExElems / n);
    for (long long r0 = 0; r0 < m; r0 += rows)
      gemm_bf16_f32_nt_rowmajor(
          static_cast<const unsigned short*>(a) + r0 * k, b, c + r0 * n,
          std::min(rows, m - r0), n, k, beta, stream);
    return;
  }
  rocblas_handle h = get_handle();
  ROCBLAS_CHECK(rocblas_set_stream(h, (hipStream_t)stream));
  const float alpha = 1.0f;
  ROCBLAS_CHECK(rocblas_gemm_ex(
      h, rocblas_operation_transpose, rocblas_operation_none,
      (rocblas_int)n, (rocblas_int)m, (rocblas_int)k, &alpha,
      b, rocblas_datatype_bf16_r, (rocblas_int)k,
      a, rocblas_datatype_bf16_r, (rocblas_int)k, &beta,
      c, rocblas_datatype_f32_r, (rocblas_int)n,
      c, rocblas_datatype_f32_r, (rocblas_int)n,
      rocblas_datatype_f32_r, rocblas_gemm_algo_standard, 0, 0));
}

// fp32 SGEMV over a row-major basis V [rows, n] (== column-major [n, rows]):
//   trans=false: y[rows] = alpha * V x[n]   + beta * y   (projection)
//   trans=true:  y[n]    = alpha * V^T x[rows] + beta * y  (reconstruction)
// The two calls form one classical Gram-Schmidt pass (Lanczos reorth).
void sgemv_rowmajor(const float* V, const float* x, float* y, long long rows,
                    long long n, bool trans, float alpha, float beta,
                    void* stream) {
  rocblas_handle h = get_handle();
  ROCBLAS_CHECK(rocblas_set_stream(h, (hipStream_t)stream));
  ROCBLAS_CHECK(rocblas_sgemv(
      h, trans ? rocblas_operation_none : rocblas_operation_transpose,
      (rocblas_int)n, (rocblas_int)rows, &alpha, V, (rocblas_int)n, x, 1,
      &beta, y, 1));
}

// fp32 SGEMM (row-major) — the native fp32 vector-ALU path for comparison.
void gemm_f32_rowmajor(const float* a, const float* b, float* c, long long m,
                       long long n, long long k, float beta, void* stream) {
  rocblas_handle h = get_handle();
  ROCBLAS_CHECK(rocblas_set_stream(h, (hipStream_t)stream));
  const float alpha = 1.0f;
  ROCBLAS_CHECK(rocblas_sgemm(h, rocblas_operation_none, rocblas_operation_none,
                              (rocblas_int)n, (rocblas_int)m, (rocblas_int)k,
                              &alpha, b, (rocblas_int)n, a, (rocblas_int)k,
                              &beta, c, (rocblas_int)n));
}

}  // namespace raft_amd
