// mdspan-taking overloads of the main raft_amd C++ entry points — the
// north-star "mdspan/mdarray C++ header API" over the compiled launchers.
//
// Reference parity: RAFT's public headers pair every raw-pointer legacy API
// with an mdspan API (e.g. matrix/select_k.cuh:75, linalg/reduce.cuh:148);
// these inline overloads play that role over libraft_amd. Device views are
// type-tagged (core/mdspan.hpp): passing a host view is a compile error.
#pragma once

#include "core/mdspan.hpp"
#include "distance.hpp"
#include "matrix.hpp"
#include "reductions.hpp"

namespace raft_amd {

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

}  // namespace raft_amd
