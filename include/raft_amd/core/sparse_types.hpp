// Sparse container views + owning matrices for the C++ mdspan API.
//
// Reference parity: raft/core/sparse_types.hpp:91 (sparse_matrix /
// sparse_structure hierarchy), core/device_csr_matrix.hpp and
// core/device_coo_matrix.hpp (owning device CSR/COO + structure views).
// MI355X design: flat aggregate views over device pointers (no CRTP tower —
// the structure/values split buys nothing here because every consumer kernel
// takes raw spans), with owning variants built on device_uvector (RAII
// hipMalloc, core/mdspan.hpp).
#pragma once

#include <stdexcept>

#include "mdspan.hpp"

namespace raft_amd {

// non-owning CSR view (device memory): y = A x consumers index as
// values[indptr[r] .. indptr[r+1]) with column ids indices[...]
template <typename T, typename IndexT = int>
struct device_csr_view {
  const IndexT* indptr;   // [n_rows + 1]
  const IndexT* indices;  // [nnz]
  const T* values;        // [nnz]
  long long n_rows = 0, n_cols = 0, nnz = 0;
};

// non-owning COO view (device memory), one (row, col, value) triple per nnz
template <typename T, typename IndexT = int>
struct device_coo_view {
  const IndexT* rows;  // [nnz]
  const IndexT* cols;  // [nnz]
  const T* values;     // [nnz] (may be nullptr for a pure structure/pattern)
  long long n_rows = 0, n_cols = 0, nnz = 0;
};

// owning CSR matrix: uninitialized device buffers; fill via copy() from host
// or device kernels, then .view() into the algorithms
template <typename T, typename IndexT = int>
class device_csr_matrix {
 public:
  device_csr_matrix(long long n_rows, long long n_cols, long long nnz)
      : n_rows_(n_rows), n_cols_(n_cols), nnz_(nnz),
        indptr_(static_cast<std::size_t>(n_rows) + 1),
        indices_(static_cast<std::size_t>(nnz)),
        values_(static_cast<std::size_t>(nnz)) {}

  device_csr_view<T, IndexT> view() const {
    return {indptr_.data(), indices_.data(), values_.data(),
            n_rows_, n_cols_, nnz_};
  }
  IndexT* indptr() { return indptr_.data(); }
  IndexT* indices() { return indices_.data(); }
  T* values() { return values_.data(); }
  long long n_rows() const { return n_rows_; }
  long long n_cols() const { return n_cols_; }
  long long nnz() const { return nnz_; }

 private:
  long long n_rows_, n_cols_, nnz_;
  device_uvector<IndexT> indptr_;
  device_uvector<IndexT> indices_;
  device_uvector<T> values_;
};

template <typename T, typename IndexT = int>
class device_coo_matrix {
 public:
  device_coo_matrix(long long n_rows, long long n_cols, long long nnz)
      : n_rows_(n_rows), n_cols_(n_cols), nnz_(nnz),
        rows_(static_cast<std::size_t>(nnz)),
        cols_(static_cast<std::size_t>(nnz)),
        values_(static_cast<std::size_t>(nnz)) {}

  device_coo_view<T, IndexT> view() const {
    return {rows_.data(), cols_.data(), values_.data(), n_rows_, n_cols_, nnz_};
  }
  IndexT* rows() { return rows_.data(); }
  IndexT* cols() { return cols_.data(); }
  T* values() { return values_.data(); }
  long long n_rows() const { return n_rows_; }
  long long n_cols() const { return n_cols_; }
  long long nnz() const { return nnz_; }

 private:
  long long n_rows_, n_cols_, nnz_;
  device_uvector<IndexT> rows_;
  device_uvector<IndexT> cols_;
  device_uvector<T> values_;
};

}  // namespace raft_amd
