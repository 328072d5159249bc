// k-means device primitives (csrc/kmeans.hip): keyed row reductions, the
// fused one-X-pass update+verify kernel, and centroid helpers.
#pragma once

#include "core.hpp"

namespace raft_amd {

// naive atomic keyed reduction (small n) — workspace holds `replicas` copies
void launch_reduce_rows_by_key(const float* x, const int* keys, float* workspace,
                               float* out, long long n_rows, long long d,
                               long long n_keys, int replicas, hipStream_t s);
// sort-based: perm/keys_sorted from an argsort of keys; atomics only at run
// boundaries; fused per-key counts
// dmin/inertia_acc (optional, nullable): accumulate sum(dmin[perm[i]]) into
// inertia_acc[0] in the same pass (the k-means inertia fold)
void launch_reduce_rows_by_key_sorted(const float* x, const int* perm,
                                      const int* keys_sorted, float* sums,
                                      float* counts, const float* dmin,
                                      float* inertia_acc, long long n_rows,
                                      long long d, hipStream_t s);
// fused one-X-pass centroid-sum accumulation + exact-fp32 verify/refine of
// rows whose split-error margin is inconclusive (the bf16x2v engine's
// update step)
// lead/tail: margin-bound constants of the split mode that produced
// dmin/dmin2 (defaults = the bf16x2v engine; bf16x1v uses 2^-7 / 2^-12 —
// see csrc/kmeans.hip l2nn_verify_repair_kernel for the derivation)
void launch_kmeans_update_verify(const float* x, const int* perm,
                                 const int* keys_sorted, const float* c,
                                 const float* xn, float* dmin, int* amin,
                                 const float* dmin2, const float* cn_max_dev,
                                 float* sums, float* counts, float* inertia_acc,
                                 long long n_rows, long long d, int n_centroids,
                                 hipStream_t s, float lead = 0x1p-13f,
                                 float tail = 0x1p-18f);
// split fp32 -> nslice bf16 slices + squared row norms in one pass
// cn_max (optional): fused max(cn) into one device float (zeroed first)
void launch_split_bf16_norms(const float* c, void* s0, void* s1, void* s2, float* cn,
                             int nslice, long long n_rows, long long d, hipStream_t s,
                             float* cn_max = nullptr);
void launch_kmeans_update_centroids(const float* sums, const float* counts, float* c,
                                    long long k, long long d, hipStream_t s);

}  // namespace raft_amd
