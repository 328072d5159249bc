// GEMM wrappers (csrc/gemm_rocblas.cpp, csrc/gemm_hipblaslt.cpp):
// bf16 x bf16 -> fp32 with fp32 accumulate — the building block of the
// split-bf16 fp32 GEMM emulation. *_lt variants route through hipBLASLt
// with a per-shape heuristic-algo cache.
#pragma once

#include "core.hpp"

namespace raft_amd {

void gemm_bf16_f32_rowmajor(const void* a, const void* b, float* c, long long m,
                            long long n, long long k, float beta, void* stream);
void gemm_bf16_f32_nt_rowmajor(const void* a, const void* b, float* c, long long m,
                               long long n, long long k, float beta, void* stream);
void gemm_f32_rowmajor(const float* a, const float* b, float* c, long long m,
                       long long n, long long k, float beta, void* stream);
void gemm_bf16_f32_rowmajor_lt(const void* a, const void* b, float* c, long long m,
                               long long n, long long k, float beta, void* stream);
void gemm_bf16_f32_nt_rowmajor_lt(const void* a, const void* b, float* c, long long m,
                                  long long n, long long k, float beta, void* stream);

}  // namespace raft_amd
