// mdspan-taking overloads of the main raft_amd C++ entry points — the
// north-star "mdspan/mdarray C++ header API" over the compiled launchers.
//
// Reference parity: RAFT's public headers pair every raw-pointer legacy API
// with an mdspan API (e.g. matrix/select_k.cuh:75, linalg/reduce.cuh:148);
// these inline overloads play that role over libraft_amd. Device views are
// type-tagged (core/mdspan.hpp): passing a host view is a compile error.
#pragma once

#include "cluster.hpp"
#include "core/mdspan.hpp"
#include "core/sparse_types.hpp"
#include "distance.hpp"
#include "matrix.hpp"
#include "neighbors.hpp"
#include "reductions.hpp"
#include "sparse.hpp"

namespace raft_amd {

// ---- fused L2 nearest-neighbor (the k-means assignment step) --------------
// x [m, d] fp32, c [n, d] fp32 -> (dmin[m], amin[m]) with PROVABLY
// fp32-exact argmin: the split-bf16 (bf16x2) MFMA engine + the in-kernel
// second-best margin test + exact-fp32 rescan/repair of inconclusive rows —
// the flagship engine, reachable from pure C++ (scratch comes from the
// caller's workspace uvector, grown on demand). Requires d % 64 == 0.
inline void fused_l2nn(device_matrix_view<const float> x,
                       device_matrix_view<const float> c,
                       device_vector_view<float> dmin,
                       device_vector_view<int> amin,
                       device_uvector<char>& workspace,
                       hipStream_t stream = nullptr) {
  const long long m = x.extent(0), d = x.extent(1), n = c.extent(0);
  if (c.extent(1) != d || dmin.extent(0) != m || amin.extent(0) != m)
    throw std::invalid_argument("fused_l2nn: extents mismatch");
  if (d % 64 != 0) throw std::invalid_argument("fused_l2nn: d % 64 != 0");
  constexpr int kSlices = 2;
  // scratch layout: x slices | c slices | xn | cn | dmin2 | cn_max
  const std::size_t xs_b = static_cast<std::size_t>(m) * d * 2;
  const std::size_t cs_b = static_cast<std::size_t>(n) * d * 2;
  const std::size_t xn_b = static_cast<std::size_t>(m) * 4;
  const std::size_t cn_b = static_cast<std::size_t>(n) * 4;
  const std::size_t need = kSlices * (xs_b + cs_b) + xn_b + cn_b + xn_b + 4;
  if (workspace.size() < need) workspace = device_uvector<char>(need);
  char* p = workspace.data();
  void* xs[3] = {p, p + xs_b, nullptr};
  p += kSlices * xs_b;
  void* cs[3] = {p, p + cs_b, nullptr};
  p += kSlices * cs_b;
  float* xn = reinterpret_cast<float*>(p);
  p += xn_b;
  float* cn = reinterpret_cast<float*>(p);
  p += cn_b;
  float* dmin2 = reinterpret_cast<float*>(p);
  p += xn_b;
  float* cn_max = reinterpret_cast<float*>(p);

  launch_split_bf16_norms(x.data_handle(), xs[0], xs[1], nullptr, xn, kSlices,
                          m, d, stream);
  launch_split_bf16_norms(c.data_handle(), cs[0], cs[1], nullptr, cn, kSlices,
                          n, d, stream);
  const void* xsl[3] = {xs[0], xs[1], nullptr};
  const void* csl[3] = {cs[0], cs[1], nullptr};
  launch_fused_l2nn_split(xsl, csl, xn, cn, dmin.data_handle(),
                          amin.data_handle(), dmin2, m, static_cast<int>(n),
                          static_cast<int>(d), kSlices, stream);
  // exact-fp32 verification/repair (needs max(cn) on device: one kMax
  // row-reduction over cn viewed as a [1, n] matrix)
  launch_reduce_rows<static_cast<int>(ReduceOpCode::kMax), float>(cn, cn_max,
                                                                  1, n, stream);
  launch_l2nn_verify_repair(x.data_handle(), c.data_handle(), xn,
                            dmin.data_handle(), amin.data_handle(), dmin2,
                            cn_max, m, static_cast<int>(n),
                            static_cast<int>(d), stream);
}

// ---- matrix::select_k -----------------------------------------------------
inline void select_k(device_matrix_view<const float> in,
                     device_matrix_view<float> out_values,
                     device_matrix_view<int> out_indices,
                     device_uvector<char>& workspace, bool select_min = true,
                     bool sorted = true, hipStream_t stream = nullptr) {
  const long long batch = in.extent(0);
  const long long len = in.extent(1);
  const int k = static_cast<int>(out_values.extent(1));
  if (out_indices.extent(0) != batch || out_values.extent(0) != batch ||
      out_indices.extent(1) != k)
    throw std::invalid_argument("select_k: output extents mismatch");
  const long long need = select_k_workspace_bytes(batch);
  if (static_cast<long long>(workspace.size()) < need)
    workspace = device_uvector<char>(static_cast<std::size_t>(need));
  launch_select_k(in.data_handle(), out_values.data_handle(),
                  out_indices.data_handle(), workspace.data(), batch, len, k,
                  select_min, sorted, stream);
}

// ---- pairwise distance (unexpanded metrics) -------------------------------
inline void pairwise_distance(device_matrix_view<const float> x,
                              device_matrix_view<const float> y,
                              device_matrix_view<float> out, DistanceCode code,
                              float p = 2.0f, hipStream_t stream = nullptr) {
  if (x.extent(1) != y.extent(1) || out.extent(0) != x.extent(0) ||
      out.extent(1) != y.extent(0))
    throw std::invalid_argument("pairwise_distance: extents mismatch");
  launch_pairwise_unexpanded(x.data_handle(), y.data_handle(),
                             out.data_handle(), x.extent(0), y.extent(0),
                             x.extent(1), static_cast<int>(code), p, stream);
}

// ---- row reductions -------------------------------------------------------
template <ReduceOpCode Op = ReduceOpCode::kSum>
inline void reduce_rows(device_matrix_view<const float> in,
                        device_vector_view<float> out,
                        hipStream_t stream = nullptr) {
  if (out.extent(0) != in.extent(0))
    throw std::invalid_argument("reduce_rows: output extent mismatch");
  launch_reduce_rows<static_cast<int>(Op), float>(
      in.data_handle(), out.data_handle(), in.extent(0), in.extent(1), stream);
}

inline void row_argmin(device_matrix_view<const float> in,
                       device_vector_view<int> out,
                       hipStream_t stream = nullptr) {
  if (out.extent(0) != in.extent(0))
    throw std::invalid_argument("row_argmin: output extent mismatch");
  launch_row_argmin(in.data_handle(), out.data_handle(), in.extent(0),
                    in.extent(1), stream);
}

// ---- sparse ---------------------------------------------------------------
// y = A x over a CSR view (reference parity: raft sparse/linalg/spmv.cuh);
// the sub-wave-per-row kernel picks 2..64 lanes/row from mean nnz/row
template <typename T>
inline void spmv(device_csr_view<T> a, device_vector_view<const T> x,
                 device_vector_view<T> y, hipStream_t stream = nullptr) {
  if (x.extent(0) != a.n_cols || y.extent(0) != a.n_rows)
    throw std::invalid_argument("spmv: extents mismatch");
  launch_csr_spmv<T>(a.indptr, a.indices, a.values, x.data_handle(),
                     y.data_handle(), a.n_rows, a.nnz, stream);
}

// SDDMM: out_vals[e] = dot(a[rows[e]], b[cols[e]]) for a COO sampling
// pattern (reference parity: raft sparse/linalg/sddmm.cuh)
inline void sddmm(device_matrix_view<const float> a,
                  device_matrix_view<const float> b,
                  device_coo_view<float> pattern,
                  device_vector_view<float> out_vals,
                  hipStream_t stream = nullptr) {
  if (a.extent(1) != b.extent(1) || out_vals.extent(0) != pattern.nnz)
    throw std::invalid_argument("sddmm: extents mismatch");
  launch_sddmm(a.data_handle(), b.data_handle(), pattern.rows, pattern.cols,
               out_vals.data_handle(), pattern.nnz, a.extent(1), stream);
}

}  // namespace raft_amd
