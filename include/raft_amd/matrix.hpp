// k-selection (csrc/select_k.hip): MSB-radix with gmem candidate workspace
// + wave-register warpsort queues, dispatched by measured shape heuristic.
#pragma once

#include "core.hpp"

namespace raft_amd {

long long select_k_workspace_bytes(long long batch);
void launch_select_k(const float* x, float* out_v, int* out_i, void* cand_workspace,
                     long long batch, long long len, int k, bool select_min,
                     bool do_sort, hipStream_t s);
// k <= 64 (wave width): per-wave register priority queue, one streaming read
void launch_select_k_warpsort(const float* x, float* out_v, int* out_i,
                              long long batch, long long len, int k,
                              bool select_min, hipStream_t s);

}  // namespace raft_amd
