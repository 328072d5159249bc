// Fused linewise (matrix x vector broadcast) engine.
//
// Reference parity: raft/matrix/detail/linewise_op.cuh:252-446 (the 5-kernel
// vectorized broadcast engine behind matrix_vector_op) — here one float4
// grid-stride kernel per orientation, with up to TWO fused vector stages so
// chains like (x - mu) / sigma run in ONE HBM pass (two torch broadcasts =
// two passes; the op is pure-bandwidth so fusion halves its cost).
//
// op codes: 0 add, 1 sub, 2 mul, 3 div.

#include <hip/hip_runtime.h>

#include "common.h"

namespace raft_amd {

__device__ __forceinline__ float lw_apply(float x, float v, int op) {
  switch (op) {
    case 0: return x + v;
    case 1: return x - v;
    case 2: return x * v;
    default: return x / v;
  }
}

// along-rows: vectors have length d, broadcast across rows (row-major [n, d])
template <int NV>
__global__ void linewise_rows_kernel(const float* __restrict__ x,
                                     float* __restrict__ out,
                                     const float* __restrict__ v1,
                                     const float* __restrict__ v2,
                                     long long n, long long d, int op1,
                                     int op2) {
  const long long total4 = n * d / 4;
  const bool d4 = (d & 3) == 0;
  if (d4) {
    const float4* x4 = reinterpret_cast<const float4*>(x);
    float4* o4 = reinterpret_cast<float4*>(out);
    const float4* v14 = reinterpret_cast<const float4*>(v1);
    const float4* v24 = reinterpret_cast<const float4*>(v2);
    const long long d_4 = d / 4;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < total4; i += (long long)gridDim.x * blockDim.x) {
      const long long c4 = i % d_4;
      float4 xv = x4[i];
      const float4 a = v14[c4];
      xv.x = lw_apply(xv.x, a.x, op1);
      xv.y = lw_apply(xv.y, a.y, op1);
      xv.z = lw_apply(xv.z, a.z, op1);
      xv.w = lw_apply(xv.w, a.w, op1);
      if (NV == 2) {
        const float4 b = v24[c4];
        xv.x = lw_apply(xv.x, b.x, op2);
        xv.y = lw_apply(xv.y, b.y, op2);
        xv.z = lw_apply(xv.z, b.z, op2);
        xv.w = lw_apply(xv.w, b.w, op2);
      }
      o4[i] = xv;
    }
    return;
  }
  const long long total = n * d;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long c = i % d;
    float r = lw_apply(x[i], v1[c], op1);
    if (NV == 2) r = lw_apply(r, v2[c], op2);
    out[i] = r;
  }
}

// along-cols: vectors have length n, broadcast across columns
template <int NV>
__global__ void linewise_cols_kernel(const float* __restrict__ x,
                                     float* __restrict__ out,
                                     const float* __restrict__ v1,
                                     const float* __restrict__ v2,
                                     long long n, long long d, int op1,
                                     int op2) {
  const bool d4 = (d & 3) == 0;
  if (d4) {
    const float4* x4 = reinterpret_cast<const float4*>(x);
    float4* o4 = reinterpret_cast<float4*>(out);
    const long long d_4 = d / 4;
    const long long total4 = n * d_4;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < total4; i += (long long)gridDim.x * blockDim.x) {
      const long long r = i / d_4;
      const float a = v1[r];
      float4 xv = x4[i];
      xv.x = lw_apply(xv.x, a, op1);
      xv.y = lw_apply(xv.y, a, op1);
      xv.z = lw_apply(xv.z, a, op1);
      xv.w = lw_apply(xv.w, a, op1);
      if (NV == 2) {
        const float b = v2[r];
        xv.x = lw_apply(xv.x, b, op2);
        xv.y = lw_apply(xv.y, b, op2);
        xv.z = lw_apply(xv.z, b, op2);
        xv.w = lw_apply(xv.w, b, op2);
      }
      o4[i] = xv;
    }
    return;
  }
  const long long total = n * d;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long r = i / d;
    float val = lw_apply(x[i], v1[r], op1);
    if (NV == 2) val = lw_apply(val, v2[r], op2);
    out[i] = val;
  }
}

void launch_linewise(const float* x, float* out, const float* v1,
                     const float* v2, long long n, long long d, bool along_rows,
                     int op1, int op2, hipStream_t stream) {
  const long long total = n * d;
  long long grid = (total / 4 + 255) / 256;
  if (grid > 4096) grid = 4096;
  if (grid < 1) grid = 1;
  if (along_rows) {
    if (v2)
      hipLaunchKernelGGL((linewise_rows_kernel<2>), dim3((unsigned)grid),
                         dim3(256), 0, stream, x, out, v1, v2, n, d, op1, op2);
    else
      hipLaunchKernelGGL((linewise_rows_kernel<1>), dim3((unsigned)grid),
                         dim3(256), 0, stream, x, out, v1, v2, n, d, op1, op2);
  } else {
    if (v2)
      hipLaunchKernelGGL((linewise_cols_kernel<2>), dim3((unsigned)grid),
                         dim3(256), 0, stream, x, out, v1, v2, n, d, op1, op2);
    else
      hipLaunchKernelGGL((linewise_cols_kernel<1>), dim3((unsigned)grid),
                         dim3(256), 0, stream, x, out, v1, v2, n, d, op1, op2);
  }
}

}  // namespace raft_amd
