// C++ resources handle: owned HIP streams (main + pool) and a grow-on-demand
// device workspace, passed by reference through the C++ API.
//
// Reference parity: raft/core/device_resources.hpp (ctor :78-92) and the
// stream-pool / workspace accessors of core/resource/*.hpp. MI355X design:
// a flat struct of owned HIP objects instead of the reference's type-indexed
// lazy registry — the registry exists to host a dozen vendor-library handles
// (cuBLAS/cuSOLVER/cuSPARSE/...); here the compiled launchers take streams
// directly and the rocBLAS handle is process-global (csrc/gemm_rocblas.cpp),
// so the handle only needs to OWN streams and scratch. The Python Resources
// (raft_amd/core/resources.py) carries the richer slot system.
#pragma once

#include <hip/hip_runtime.h>

#include <vector>

#include "mdspan.hpp"

namespace raft_amd {

class device_resources {
 public:
  explicit device_resources(int device_id = 0, int n_pool_streams = 0)
      : device_(device_id) {
    int prev = 0;
    check_hip_(hipGetDevice(&prev), "hipGetDevice");
    check_hip_(hipSetDevice(device_), "hipSetDevice");
    check_hip_(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking),
               "hipStreamCreate");
    pool_.resize(n_pool_streams);
    for (auto& s : pool_)
      check_hip_(hipStreamCreateWithFlags(&s, hipStreamNonBlocking),
                 "hipStreamCreate(pool)");
    check_hip_(hipSetDevice(prev), "hipSetDevice(restore)");
  }
  device_resources(const device_resources&) = delete;
  device_resources& operator=(const device_resources&) = delete;
  device_resources(device_resources&& o) noexcept
      : device_(o.device_), stream_(o.stream_), pool_(std::move(o.pool_)),
        workspace_(std::move(o.workspace_)) {
    o.stream_ = nullptr;
    o.pool_.clear();
  }
  ~device_resources() {
    if (stream_) (void)hipStreamDestroy(stream_);
    for (auto s : pool_)
      if (s) (void)hipStreamDestroy(s);
  }

  int get_device() const noexcept { return device_; }
  hipStream_t get_stream() const noexcept { return stream_; }
  std::size_t stream_pool_size() const noexcept { return pool_.size(); }
  // round-robin access (device_resources_manager parity: callers index by
  // worker id; modulo keeps any index valid)
  hipStream_t get_stream_from_pool(std::size_t i) const {
    return pool_.empty() ? stream_ : pool_[i % pool_.size()];
  }
  void sync_stream() const {
    check_hip_(hipStreamSynchronize(stream_), "hipStreamSynchronize");
  }
  void sync_stream_pool() const {
    for (auto s : pool_)
      check_hip_(hipStreamSynchronize(s), "hipStreamSynchronize(pool)");
  }

  // grow-on-demand scratch shared by the mdspan-API entry points (e.g.
  // select_k / fused_l2nn take a device_uvector<char>& workspace)
  device_uvector<char>& get_workspace() { return workspace_; }

 private:
  int device_ = 0;
  hipStream_t stream_ = nullptr;
  std::vector<hipStream_t> pool_;
  device_uvector<char> workspace_;
};

}  // namespace raft_amd
