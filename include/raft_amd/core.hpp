// raft_amd C++ core: common types for the public kernel API.
// Reference parity: raft_runtime's non-templated entry-point idea.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace raft_amd {

// reduction op codes shared by reduce_rows/reduce_cols
// (keep in sync with raft_amd/linalg/reduce.py _EXT_CODES)
enum class ReduceOpCode : int {
  kSum = 0, kSumSq = 1, kSumAbs = 2, kMax = 3, kMin = 4, kMaxAbs = 5,
};

// unexpanded pairwise-distance codes (launch_pairwise_unexpanded)
enum class DistanceCode : int {
  kL1 = 0, kLinf = 1, kLp = 2, kCanberra = 3, kHamming = 4,
};

}  // namespace raft_amd
