// raft_amd — public C++ API of the native MI355X (gfx950) kernel library.
//
// These are the host-callable entry points implemented in csrc/ and compiled
// into the raft_amd extension; they are plain C++ functions over device
// pointers + a hipStream_t, usable from any C++ host code linked against the
// objects in build/ext (no Python required). The Python package binds them
// 1:1 (csrc/bindings.cpp).
//
// Reference parity: the raft_runtime precompiled-API idea
// (cpp/include/raft_runtime in rapidsai/raft) — non-templated entry points a
// consumer can call without a device compiler.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace raft_amd {

// -- reductions (csrc/reductions.hip) ---------------------------------------
// op codes: 0 sum(x), 1 sum(x^2), 2 sum(|x|), 3 max, 4 min, 5 max(|x|)
template <int OP, typename T>
void launch_reduce_rows(const T* x, T* out, long long n_rows, long long d, hipStream_t s);
template <int OP, typename T>
void launch_reduce_cols(const T* x, T* out, long long n_rows, long long d, hipStream_t s);
void launch_row_argmin(const float* x, int* out, long long n_rows, long long d, hipStream_t s);
void launch_row_normalize_l2(const float* x, float* out, long long n_rows, long long d,
                             float eps, hipStream_t s);

// -- pairwise distances (csrc/pairwise.hip, csrc/pairwise_mfma.hip) ---------
void launch_l2_epilogue(float* g, const float* xn, const float* yn, long long m,
                        long long n, hipStream_t s);
void launch_l2nn_epilogue(const float* g, const float* xn, const float* yn,
                          float* dmin, int* amin, long long m, long long n, hipStream_t s);
// codes: 0 L1, 1 Linf, 2 Lp, 3 Canberra, 4 Hamming
void launch_pairwise_unexpanded(const float* x, const float* y, float* out, long long m,
                                long long n, long long d, int code, float p, hipStream_t s);
// split-bf16 MFMA tile kernel: xsl/csl = nslice bf16 slice pointers
void launch_pairwise_l2_mfma(const void** xsl, const void** csl, const float* xn,
                             const float* yn, float* out, long long m, long long n,
                             int d, long long ldo, int nslice, bool sqrt_out, hipStream_t s);

// -- fused L2 nearest neighbor (csrc/fused_l2nn.hip) ------------------------
void launch_fused_l2nn_split(const void** xsl, const void** csl, const float* xn,
                             const float* cn, float* dmin, int* amin, float* dmin2,
                             long long m, int n, int d, int nslice, hipStream_t s);
void launch_l2nn_verify_repair(const float* x, const float* c, const float* xn,
                               float* dmin, int* amin, const float* dmin2, float cn_max,
                               long long m, int n, int d, hipStream_t s);

// -- k-selection (csrc/select_k.hip) ----------------------------------------
long long select_k_workspace_bytes(long long batch);
void launch_select_k(const float* x, float* out_v, int* out_i, void* cand_workspace,
                     long long batch, long long len, int k, bool select_min,
                     bool do_sort, hipStream_t s);

// -- RNG + generators (csrc/rng.hip) ----------------------------------------
void launch_rng_uniform(float* out, long long n, uint64_t seed, uint64_t subseq, hipStream_t s);
void launch_rng_normal(float* out, long long n, uint64_t seed, uint64_t subseq, hipStream_t s);
void launch_make_blobs(float* x, int* labels, const float* centers, long long n_rows,
                       long long d, int k, float cluster_std, uint64_t seed,
                       uint64_t subseq, hipStream_t s);

// -- sparse (csrc/spmv.hip) --------------------------------------------------
template <typename T>
void launch_csr_spmv(const int* indptr, const int* indices, const T* values, const T* x,
                     T* y, long long n_rows, long long nnz, hipStream_t s);

// -- keyed reductions (csrc/kmeans.hip) -------------------------------------
void launch_reduce_rows_by_key(const float* x, const int* keys, float* workspace,
                               float* out, long long n_rows, long long d,
                               long long n_keys, int replicas, hipStream_t s);
void launch_reduce_rows_by_key_sorted(const float* x, const int* perm,
                                      const int* keys_sorted, float* sums,
                                      long long n_rows, long long d, hipStream_t s);

// -- GEMM wrappers (csrc/gemm_rocblas.cpp) ----------------------------------
void gemm_bf16_f32_rowmajor(const void* a, const void* b, float* c, long long m,
                            long long n, long long k, float beta, void* stream);
void gemm_bf16_f32_nt_rowmajor(const void* a, const void* b, float* c, long long m,
                               long long n, long long k, float beta, void* stream);
void gemm_f32_rowmajor(const float* a, const float* b, float* c, long long m,
                       long long n, long long k, float beta, void* stream);

}  // namespace raft_amd
