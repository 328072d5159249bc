// raft_amd mdspan/mdarray — the header-only multi-dimensional view and
// owning-array API of the C++ surface.
//
// Reference parity: raft/core/device_mdspan.hpp:27-98 (memory-type-tagged
// view aliases), core/mdarray.hpp:123 (owning array over a container
// policy), device_mdarray.hpp:127-183 (factory functions), core/span.
// The reference rides CCCL's std::experimental::mdspan; this is a
// self-contained C++17 implementation of the subset RAFT's API uses:
// static/dynamic extents, layout_right/layout_left/layout_stride, accessor
// policies carrying the MEMORY TYPE (host/device views are distinct types,
// so passing a host view to a device entry point is a compile error — the
// same safety the reference gets from its accessor mixins).
//
// Owning arrays: device_uvector<T> (RAII hipMalloc, the rmm::device_uvector
// analog) and device_mdarray / host_mdarray with .view() accessors.
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <cstdint>
#include <memory>
#include <stdexcept>
#include <type_traits>
#include <utility>
#include <vector>

namespace raft_amd {

inline constexpr std::size_t dynamic_extent = static_cast<std::size_t>(-1);

// ---------------------------------------------------------------------------
// extents<IndexType, E0, E1, ...> — static/dynamic mix
// ---------------------------------------------------------------------------
template <class IndexType, std::size_t... StaticExtents>
class extents {
 public:
  using index_type = IndexType;
  static constexpr std::size_t rank() noexcept { return sizeof...(StaticExtents); }

  static constexpr std::size_t rank_dynamic() noexcept {
    std::size_t n = 0;
    for (auto e : kStatic) n += (e == dynamic_extent);
    return n;
  }

  constexpr extents() noexcept = default;

  template <class... DynSizes,
            std::enable_if_t<sizeof...(DynSizes) == sizeof...(StaticExtents) ||
                                 sizeof...(DynSizes) == 0,
                             int> = 0>
  constexpr explicit extents(DynSizes... dyn) noexcept {
    if constexpr (sizeof...(DynSizes) == sizeof...(StaticExtents)) {
      const IndexType vals[] = {static_cast<IndexType>(dyn)...};
      for (std::size_t i = 0; i < rank(); i++)
        ext_[i] = kStatic[i] == dynamic_extent ? vals[i]
                                               : static_cast<IndexType>(kStatic[i]);
    }
  }

  static constexpr std::size_t static_extent(std::size_t i) noexcept {
    return kStatic[i];
  }
  constexpr IndexType extent(std::size_t i) const noexcept { return ext_[i]; }

  constexpr bool operator==(const extents& o) const noexcept {
    for (std::size_t i = 0; i < rank(); i++)
      if (ext_[i] != o.ext_[i]) return false;
    return true;
  }

 private:
  static constexpr std::size_t kStatic[sizeof...(StaticExtents)] = {StaticExtents...};
  IndexType ext_[sizeof...(StaticExtents)] = {};
};

namespace detail {
template <class IndexType, class Seq>
struct dextents_impl;
template <class IndexType, std::size_t... Is>
struct dextents_impl<IndexType, std::index_sequence<Is...>> {
  template <std::size_t>
  static constexpr std::size_t dyn() { return dynamic_extent; }
  using type = extents<IndexType, ((void)Is, dynamic_extent)...>;
};
}  // namespace detail

template <class IndexType, std::size_t Rank>
using dextents =
    typename detail::dextents_impl<IndexType,
                                   std::make_index_sequence<Rank>>::type;

// ---------------------------------------------------------------------------
// layouts
// ---------------------------------------------------------------------------
struct layout_right {  // row-major (C order) — the default everywhere
  template <class Extents>
  class mapping {
   public:
    using extents_type = Extents;
    using index_type = typename Extents::index_type;
    constexpr mapping() noexcept = default;
    constexpr explicit mapping(const Extents& e) noexcept : ext_(e) {}
    constexpr const Extents& extents() const noexcept { return ext_; }
    template <class... Idx>
    constexpr index_type operator()(Idx... idx) const noexcept {
      static_assert(sizeof...(Idx) == Extents::rank(), "index rank mismatch");
      const index_type is[] = {static_cast<index_type>(idx)...};
      index_type off = 0;
      for (std::size_t i = 0; i < Extents::rank(); i++)
        off = off * ext_.extent(i) + is[i];
      return off;
    }
    constexpr index_type required_span_size() const noexcept {
      index_type s = 1;
      for (std::size_t i = 0; i < Extents::rank(); i++) s *= ext_.extent(i);
      return s;
    }
    constexpr index_type stride(std::size_t r) const noexcept {
      index_type s = 1;
      for (std::size_t i = Extents::rank(); i-- > r + 1;) s *= ext_.extent(i);
      return s;
    }
   private:
    Extents ext_{};
  };
};

struct layout_left {  // column-major (Fortran order)
  template <class Extents>
  class mapping {
   public:
    using extents_type = Extents;
    using index_type = typename Extents::index_type;
    constexpr mapping() noexcept = default;
    constexpr explicit mapping(const Extents& e) noexcept : ext_(e) {}
    constexpr const Extents& extents() const noexcept { return ext_; }
    template <class... Idx>
    constexpr index_type operator()(Idx... idx) const noexcept {
      static_assert(sizeof...(Idx) == Extents::rank(), "index rank mismatch");
      const index_type is[] = {static_cast<index_type>(idx)...};
      index_type off = 0;
      for (std::size_t i = Extents::rank(); i-- > 0;)
        off = off * ext_.extent(i) + is[i];
      return off;
    }
    constexpr index_type required_span_size() const noexcept {
      index_type s = 1;
      for (std::size_t i = 0; i < Extents::rank(); i++) s *= ext_.extent(i);
      return s;
    }
    constexpr index_type stride(std::size_t r) const noexcept {
      index_type s = 1;
      for (std::size_t i = 0; i < r; i++) s *= ext_.extent(i);
      return s;
    }
   private:
    Extents ext_{};
  };
};

struct layout_stride {  // explicit strides (padded / sliced views)
  template <class Extents>
  class mapping {
   public:
    using extents_type = Extents;
    using index_type = typename Extents::index_type;
    constexpr mapping() noexcept = default;
    constexpr mapping(const Extents& e,
                      const index_type (&strides)[Extents::rank()]) noexcept
        : ext_(e) {
      for (std::size_t i = 0; i < Extents::rank(); i++) str_[i] = strides[i];
    }
    constexpr const Extents& extents() const noexcept { return ext_; }
    template <class... Idx>
    constexpr index_type operator()(Idx... idx) const noexcept {
      const index_type is[] = {static_cast<index_type>(idx)...};
      index_type off = 0;
      for (std::size_t i = 0; i < Extents::rank(); i++) off += is[i] * str_[i];
      return off;
    }
    constexpr index_type required_span_size() const noexcept {
      index_type s = 1;
      for (std::size_t i = 0; i < Extents::rank(); i++)
        s += (ext_.extent(i) - 1) * str_[i];
      return s;
    }
    constexpr index_type stride(std::size_t r) const noexcept { return str_[r]; }
   private:
    Extents ext_{};
    index_type str_[Extents::rank()] = {};
  };
};

// ---------------------------------------------------------------------------
// memory-type-tagged accessors (device/host views are DISTINCT types)
// ---------------------------------------------------------------------------
enum class memory_kind { host, device, managed };

template <class T, memory_kind Kind>
struct tagged_accessor {
  using element_type = T;
  using reference = T&;
  using data_handle_type = T*;
  static constexpr memory_kind kind = Kind;
  constexpr reference access(data_handle_type p, std::size_t i) const noexcept {
    return p[i];
  }
  constexpr data_handle_type offset(data_handle_type p,
                                    std::size_t i) const noexcept {
    return p + i;
  }
};

template <class T>
using device_accessor = tagged_accessor<T, memory_kind::device>;
template <class T>
using host_accessor = tagged_accessor<T, memory_kind::host>;
template <class T>
using managed_accessor = tagged_accessor<T, memory_kind::managed>;

// ---------------------------------------------------------------------------
// mdspan
// ---------------------------------------------------------------------------
template <class T, class Extents, class Layout = layout_right,
          class Accessor = device_accessor<T>>
class mdspan {
 public:
  using element_type = T;
  using extents_type = Extents;
  using layout_type = Layout;
  using accessor_type = Accessor;
  using mapping_type = typename Layout::template mapping<Extents>;
  using index_type = typename Extents::index_type;
  using data_handle_type = typename Accessor::data_handle_type;

  constexpr mdspan() noexcept = default;
  constexpr mdspan(data_handle_type p, const Extents& e) noexcept
      : ptr_(p), map_(e) {}
  constexpr mdspan(data_handle_type p, const mapping_type& m) noexcept
      : ptr_(p), map_(m) {}

  template <class... Idx>
  constexpr typename Accessor::reference operator()(Idx... idx) const noexcept {
    return acc_.access(ptr_, static_cast<std::size_t>(map_(idx...)));
  }
  constexpr data_handle_type data_handle() const noexcept { return ptr_; }
  constexpr const mapping_type& mapping() const noexcept { return map_; }
  constexpr const Extents& extents() const noexcept { return map_.extents(); }
  constexpr index_type extent(std::size_t i) const noexcept {
    return map_.extents().extent(i);
  }
  constexpr index_type stride(std::size_t i) const noexcept {
    return map_.stride(i);
  }
  static constexpr std::size_t rank() noexcept { return Extents::rank(); }
  constexpr std::size_t size() const noexcept {
    std::size_t s = 1;
    for (std::size_t i = 0; i < rank(); i++)
      s *= static_cast<std::size_t>(extent(i));
    return s;
  }
  static constexpr memory_kind kind() noexcept { return Accessor::kind; }

 private:
  data_handle_type ptr_ = nullptr;
  mapping_type map_{};
  [[no_unique_address]] Accessor acc_{};
};

// -- RAFT-shaped aliases (device_mdspan.hpp:27-98 parity) -------------------
template <class T, class IndexType = std::int64_t, class Layout = layout_right>
using device_matrix_view =
    mdspan<T, dextents<IndexType, 2>, Layout, device_accessor<T>>;
template <class T, class IndexType = std::int64_t>
using device_vector_view =
    mdspan<T, dextents<IndexType, 1>, layout_right, device_accessor<T>>;
template <class T, class IndexType = std::int64_t>
using device_scalar_view =
    mdspan<T, extents<IndexType>, layout_right, device_accessor<T>>;
template <class T, class IndexType = std::int64_t, class Layout = layout_right>
using host_matrix_view =
    mdspan<T, dextents<IndexType, 2>, Layout, host_accessor<T>>;
template <class T, class IndexType = std::int64_t>
using host_vector_view =
    mdspan<T, dextents<IndexType, 1>, layout_right, host_accessor<T>>;

template <class T, class IndexType = std::int64_t>
constexpr auto make_device_matrix_view(T* p, IndexType rows, IndexType cols) {
  return device_matrix_view<T, IndexType>(p, dextents<IndexType, 2>(rows, cols));
}
template <class T, class IndexType = std::int64_t>
constexpr auto make_device_vector_view(T* p, IndexType n) {
  return device_vector_view<T, IndexType>(p, dextents<IndexType, 1>(n));
}
template <class T, class IndexType = std::int64_t>
constexpr auto make_host_matrix_view(T* p, IndexType rows, IndexType cols) {
  return host_matrix_view<T, IndexType>(p, dextents<IndexType, 2>(rows, cols));
}
template <class T, class IndexType = std::int64_t>
constexpr auto make_host_vector_view(T* p, IndexType n) {
  return host_vector_view<T, IndexType>(p, dextents<IndexType, 1>(n));
}

// ---------------------------------------------------------------------------
// owning arrays
// ---------------------------------------------------------------------------
inline void check_hip_(hipError_t e, const char* what) {
  if (e != hipSuccess) throw std::runtime_error(std::string(what) + ": " +
                                                hipGetErrorString(e));
}

// rmm::device_uvector analog: RAII device buffer (uninitialized)
template <class T>
class device_uvector {
 public:
  device_uvector() noexcept = default;
  explicit device_uvector(std::size_t n) : n_(n) {
    if (n_) check_hip_(hipMalloc(&p_, n_ * sizeof(T)), "hipMalloc");
  }
  device_uvector(device_uvector&& o) noexcept : p_(o.p_), n_(o.n_) {
    o.p_ = nullptr;
    o.n_ = 0;
  }
  device_uvector& operator=(device_uvector&& o) noexcept {
    if (this != &o) {
      free_();
      p_ = o.p_;
      n_ = o.n_;
      o.p_ = nullptr;
      o.n_ = 0;
    }
    return *this;
  }
  device_uvector(const device_uvector&) = delete;
  device_uvector& operator=(const device_uvector&) = delete;
  ~device_uvector() { free_(); }
  T* data() noexcept { return p_; }
  const T* data() const noexcept { return p_; }
  std::size_t size() const noexcept { return n_; }
 private:
  void free_() noexcept {
    if (p_) (void)hipFree(p_);
    p_ = nullptr;
  }
  T* p_ = nullptr;
  std::size_t n_ = 0;
};

// mdarray (owning) over a device_uvector / std::vector container policy
template <class T, class Extents, class Layout = layout_right>
class device_mdarray {
 public:
  using mapping_type = typename Layout::template mapping<Extents>;
  using view_type = mdspan<T, Extents, Layout, device_accessor<T>>;
  using const_view_type = mdspan<const T, Extents, Layout, device_accessor<const T>>;
  explicit device_mdarray(const Extents& e)
      : map_(e), buf_(static_cast<std::size_t>(map_.required_span_size())) {}
  view_type view() noexcept { return view_type(buf_.data(), map_); }
  const_view_type view() const noexcept {
    return const_view_type(buf_.data(), map_);
  }
  T* data_handle() noexcept { return buf_.data(); }
  const T* data_handle() const noexcept { return buf_.data(); }
  std::size_t size() const noexcept { return buf_.size(); }
 private:
  mapping_type map_;
  device_uvector<T> buf_;
};

template <class T, class Extents, class Layout = layout_right>
class host_mdarray {
 public:
  using mapping_type = typename Layout::template mapping<Extents>;
  using view_type = mdspan<T, Extents, Layout, host_accessor<T>>;
  explicit host_mdarray(const Extents& e)
      : map_(e), buf_(static_cast<std::size_t>(map_.required_span_size())) {}
  view_type view() noexcept { return view_type(buf_.data(), map_); }
  T* data_handle() noexcept { return buf_.data(); }
  std::size_t size() const noexcept { return buf_.size(); }
 private:
  mapping_type map_;
  std::vector<T> buf_;
};

// factories (device_mdarray.hpp:127-183 parity)
template <class T, class IndexType = std::int64_t>
auto make_device_matrix(IndexType rows, IndexType cols) {
  return device_mdarray<T, dextents<IndexType, 2>>(
      dextents<IndexType, 2>(rows, cols));
}
template <class T, class IndexType = std::int64_t>
auto make_device_vector(IndexType n) {
  return device_mdarray<T, dextents<IndexType, 1>>(dextents<IndexType, 1>(n));
}
template <class T, class IndexType = std::int64_t>
auto make_host_matrix(IndexType rows, IndexType cols) {
  return host_mdarray<T, dextents<IndexType, 2>>(
      dextents<IndexType, 2>(rows, cols));
}
template <class T, class IndexType = std::int64_t>
auto make_host_vector(IndexType n) {
  return host_mdarray<T, dextents<IndexType, 1>>(dextents<IndexType, 1>(n));
}

// ---------------------------------------------------------------------------
// mdbuffer — location-polymorphic owning buffer (core/mdbuffer.cuh parity):
// holds EITHER a host or a device mdarray and serves views in a REQUESTED
// memory kind, copying lazily on first cross-location request. The C++
// analog of the reference's std::variant machinery, sized to the two
// locations raft_amd uses.
// ---------------------------------------------------------------------------
template <class T, class Extents, class Layout = layout_right>
class mdbuffer {
 public:
  explicit mdbuffer(const Extents& e, memory_kind where = memory_kind::device)
      : ext_(e), kind_(where) {
    if (where == memory_kind::device)
      dev_ = std::make_unique<device_mdarray<T, Extents, Layout>>(e);
    else
      host_ = std::make_unique<host_mdarray<T, Extents, Layout>>(e);
  }

  memory_kind kind() const noexcept { return kind_; }
  const Extents& extents() const noexcept { return ext_; }

  // view in the requested location; copies across lazily (both copies stay
  // alive afterwards — the caller owns coherence, like the reference)
  mdspan<T, Extents, Layout, device_accessor<T>> device_view(
      hipStream_t stream = nullptr) {
    if (!dev_) {
      dev_ = std::make_unique<device_mdarray<T, Extents, Layout>>(ext_);
      check_hip_(hipMemcpyAsync(dev_->data_handle(), host_->data_handle(),
                                dev_->size() * sizeof(T),
                                hipMemcpyHostToDevice, stream),
                 "mdbuffer H2D");
    }
    return dev_->view();
  }
  mdspan<T, Extents, Layout, host_accessor<T>> host_view(
      hipStream_t stream = nullptr) {
    if (!host_) {
      host_ = std::make_unique<host_mdarray<T, Extents, Layout>>(ext_);
      check_hip_(hipMemcpyAsync(host_->data_handle(), dev_->data_handle(),
                                host_->size() * sizeof(T),
                                hipMemcpyDeviceToHost, stream),
                 "mdbuffer D2H");
      check_hip_(hipStreamSynchronize(stream), "mdbuffer D2H sync");
    }
    return host_->view();
  }

 private:
  Extents ext_;
  memory_kind kind_;
  std::unique_ptr<device_mdarray<T, Extents, Layout>> dev_;
  std::unique_ptr<host_mdarray<T, Extents, Layout>> host_;
};

// host<->device copies for mdarray/mdspan pairs (contiguous layouts)
template <class T, class E, class L>
void copy(mdspan<T, E, L, device_accessor<T>> dst,
          mdspan<const T, E, L, host_accessor<const T>> src,
          hipStream_t stream = nullptr) {
  check_hip_(hipMemcpyAsync(dst.data_handle(), src.data_handle(),
                            dst.size() * sizeof(T), hipMemcpyHostToDevice,
                            stream),
             "hipMemcpyAsync H2D");
}
template <class T, class E, class L>
void copy(mdspan<T, E, L, host_accessor<T>> dst,
          mdspan<const T, E, L, device_accessor<const T>> src,
          hipStream_t stream = nullptr) {
  check_hip_(hipMemcpyAsync(dst.data_handle(), src.data_handle(),
                            dst.size() * sizeof(T), hipMemcpyDeviceToHost,
                            stream),
             "hipMemcpyAsync D2H");
}

}  // namespace raft_amd
