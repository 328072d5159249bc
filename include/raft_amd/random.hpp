// Counter-based PCG32 RNG + fused generators (csrc/rng.hip) —
// bitwise-identical to the CPU torch implementation in raft_amd/random.
#pragma once

#include "core.hpp"

namespace raft_amd {

// gen: 0 = PCG32 (default), 1 = Philox4x32-10
void launch_rng_uniform(float* out, long long n, uint64_t seed, uint64_t subseq, hipStream_t s,
                        int gen = 0);
void launch_rng_normal(float* out, long long n, uint64_t seed, uint64_t subseq, hipStream_t s,
                       int gen = 0);
void launch_make_blobs(float* x, int* labels, const float* centers, long long n_rows,
                       long long d, int k, float cluster_std, uint64_t seed,
                       uint64_t subseq, hipStream_t s);

}  // namespace raft_amd
