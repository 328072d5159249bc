// Host-side semantics test of the mdspan/mdarray header (no GPU needed):
// extents, layouts, strides, host mdarray round-trip, memory-kind tagging.
#include <raft_amd/core/mdspan.hpp>

#include <cassert>
#include <cstdio>
#include <type_traits>

using namespace raft_amd;

int main() {
  // extents: static/dynamic mix
  extents<int, 3, dynamic_extent> e(3, 7);
  assert(e.rank() == 2 && e.rank_dynamic() == 1);
  assert(e.extent(0) == 3 && e.extent(1) == 7);
  static_assert(extents<int, 3, dynamic_extent>::static_extent(0) == 3);

  // layout_right strides + indexing
  dextents<std::int64_t, 3> e3(2, 3, 4);
  layout_right::mapping<dextents<std::int64_t, 3>> mr(e3);
  assert(mr(1, 2, 3) == 1 * 12 + 2 * 4 + 3);
  assert(mr.stride(0) == 12 && mr.stride(1) == 4 && mr.stride(2) == 1);
  assert(mr.required_span_size() == 24);

  // layout_left
  layout_left::mapping<dextents<std::int64_t, 3>> ml(e3);
  assert(ml(1, 2, 3) == 1 + 2 * 2 + 3 * 6);
  assert(ml.stride(0) == 1 && ml.stride(1) == 2 && ml.stride(2) == 6);

  // layout_stride (padded rows)
  const std::int64_t strides[2] = {10, 1};
  layout_stride::mapping<dextents<std::int64_t, 2>> ms(
      dextents<std::int64_t, 2>(3, 4), strides);
  assert(ms(2, 3) == 23 && ms.required_span_size() == 24);

  // host mdarray round-trip through a view
  auto h = make_host_matrix<float>(4, 5);
  auto v = h.view();
  for (int i = 0; i < 4; i++)
    for (int j = 0; j < 5; j++) v(i, j) = float(i * 10 + j);
  assert(v(3, 4) == 34.f && v.extent(0) == 4 && v.size() == 20);
  assert(h.data_handle()[3 * 5 + 4] == 34.f);

  // memory-kind tagging: host and device views are DIFFERENT types
  static_assert(!std::is_same_v<host_matrix_view<float>,
                                device_matrix_view<float>>);
  static_assert(device_matrix_view<float>::kind() == memory_kind::device);

  // mdbuffer: host-only round-trip (device paths exercised on GPU)
  mdbuffer<float, dextents<std::int64_t, 2>> buf(
      dextents<std::int64_t, 2>(3, 3), memory_kind::host);
  auto bv = buf.host_view();
  bv(2, 2) = 9.f;
  assert(buf.host_view()(2, 2) == 9.f);
  assert(buf.kind() == memory_kind::host);

  std::printf("MDSPAN_HOST_OK\n");
  return 0;
}
