// Python bindings for the raft_amd native extension (raft_amd._C).
// Torch tensors in, torch tensors out; every launch goes onto the current
// PyTorch HIP stream so the ops compose with torch's own kernels.
// Reference parity: the raft_runtime instantiation layer + pylibraft's
// Cython bridges (cpp/src/*, python/pylibraft/**/*.pyx) collapsed into one
// torch-extension TU.

#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

namespace raft_amd {

// from rng.hip (device sampling test driver)
void launch_device_sample_test(const float*, int*, int, uint64_t, hipStream_t);
// from spmv.hip (sddmm)
void launch_sddmm(const float*, const float*, const int*, const int*, float*,
                  long long, long long, hipStream_t);
// from linewise.hip
void launch_linewise(const float*, float*, const float*, const float*,
                     long long, long long, bool, int, int, hipStream_t);
// from histogram.hip
void launch_histogram(const float*, long long, long long, int, float, float,
                      unsigned long long*, hipStream_t);
void launch_bitset_set(unsigned int*, const long long*, long long, int, hipStream_t);
void launch_bitset_test(const unsigned int*, const long long*, bool*, long long,
                        hipStream_t);
void launch_bitset_count(const unsigned int*, long long, unsigned long long*,
                         hipStream_t);
// from solver_kernels.hip
void launch_cholesky_r1_update_f32(float*, float*, int, long long, hipStream_t);
void launch_lanczos_pre(float*, const float*, const float*, const float*,
                        double*, long long, hipStream_t);
void launch_lanczos_sub_alpha(float*, const float*, const double*, float*,
                              long long, hipStream_t);
void launch_lanczos_norm2(const float*, double*, long long, hipStream_t);
void sgemv_rowmajor(const float*, const float*, float*, long long, long long,
                    bool, float, float, void*);
void launch_lanczos_normalize(const float*, float*, const double*, float*,
                              float*, float*, long long, hipStream_t);
void launch_cholesky_r1_update_f64(double*, double*, int, long long, hipStream_t);
// from reductions.hip
template <int OP, typename T>
void launch_reduce_rows(const T*, T*, long long, long long, hipStream_t);
template <int OP, typename T>
void launch_reduce_cols(const T*, T*, long long, long long, hipStream_t);
void launch_row_argmin(const float*, int*, long long, long long, hipStream_t);
void launch_rows_sqnorm_bf16(const void*, float*, long long, long long, hipStream_t);
void launch_row_normalize_l2(const float*, float*, long long, long long, float, hipStream_t);
// from pairwise.hip
void launch_l2_epilogue(float*, const float*, const float*, long long, long long, hipStream_t);
void launch_l2nn_epilogue(const float*, const float*, const float*, float*, int*,
                          long long, long long, hipStream_t);
void launch_pairwise_unexpanded(const float*, const float*, float*, long long, long long,
                                long long, int, float, hipStream_t);
// from rng.hip
void launch_rng_uniform(float*, long long, uint64_t, uint64_t, hipStream_t, int);
void launch_rng_normal(float*, long long, uint64_t, uint64_t, hipStream_t, int);
void launch_make_blobs(float*, int*, const float*, long long, long long, int, float,
                       uint64_t, uint64_t, hipStream_t);
// from spmv.hip
template <typename T>
void launch_csr_spmv(const int*, const int*, const T*, const T*, T*, long long,
                     long long, hipStream_t);
// from kmeans.hip
void launch_reduce_rows_by_key(const float*, const int*, float*, float*, long long,
                               long long, long long, int, hipStream_t);
void launch_reduce_rows_by_key_sorted(const float*, const int*, const int*, float*,
                                      float*, const float*, float*, long long,
                                      long long, hipStream_t);
void launch_split_bf16_norms(const float*, void*, void*, void*, float*, int,
                             long long, long long, hipStream_t,
                             float* = nullptr);
void launch_kmeans_update_centroids(const float*, const float*, float*, long long,
                                    long long, hipStream_t);
void launch_l2nn_verify_repair(const float*, const float*, const float*, float*, int*,
                               const float*, const float*, long long, int, int,
                               hipStream_t, float, float);
void launch_kmeans_update_verify(const float*, const int*, const int*, const float*,
                                 const float*, float*, int*, const float*,
                                 const float*, float*, float*, float*, long long,
                                 long long, int, hipStream_t, float, float);
// from select_k.hip
long long select_k_workspace_bytes(long long batch);
void launch_select_k(const float*, float*, int*, void*, long long, long long, int,
                     bool, bool, hipStream_t);
void launch_select_k_warpsort(const float*, float*, int*, long long, long long, int,
                              bool, hipStream_t);
template <typename T>
void launch_select_k_generic_t(const T*, const long long*, long long, long long,
                               T*, long long*, long long, long long, bool,
                               hipStream_t);
// from fused_l2nn.hip
void launch_fused_l2nn_split(const void**, const void**, const float*, const float*,
                             float*, int*, float*, long long, int, int, int,
                             hipStream_t);
// from fused_l2nn_2d.hip (XCD-swizzled tile-pair grid + partials combine)
bool fused_l2nn_2d_supported(int nslice, long long m, int n, int d);
void launch_fused_l2nn_2d(const void**, const void**, const float*, const float*,
                          float*, float*, int*, float*, int*, float*,
                          long long, int, int, int, hipStream_t);
// from fused_l2nn_256.hip (256^2 counted-vmcnt double-buffered engine)
bool fused_l2nn_256_supported(int nslice, long long m, int n, int d);
void launch_fused_l2nn_256(const void**, const void**, const float*, const float*,
                           float*, float*, int*, float*, int*, float*,
                           long long, int, int, int, hipStream_t);
// from fused_l2nn_v2.hip (persistent-X variant + w8 wide-tile variant)
bool fused_l2nn_persist_supported(int nslice, int d);
bool fused_l2nn_w8_supported(int nslice, int n, int d);
void launch_fused_l2nn_w8(const void**, const void**, const float*, const float*,
                          float*, int*, float*, long long, int, int, int,
                          hipStream_t);
void launch_fused_l2nn_persist(const void**, const void**, const float*, const float*,
                               float*, int*, float*, long long, int, int, int,
                               hipStream_t);
// from pairwise_mfma.hip
void launch_pairwise_l2_mfma(const void**, const void**, const float*, const float*,
                             float*, long long, long long, int, long long, int, bool,
                             hipStream_t);
void launch_pairwise_l2_mfma256(const void**, const void**, const float*, const float*,
                                float*, long long, long long, int, long long, int, bool,
                                hipStream_t);
void launch_pairwise_l2_filter(const void**, const void**, const float*, const float*,
                               const float*, float*, int*, int*, int, long long,
                               long long, long long, int, int, hipStream_t);
void launch_pairwise_l2_filter256(const void**, const void**, const float*, const float*,
                                  const float*, float*, int*, int*, int, long long,
                                  long long, long long, int, int, hipStream_t);
// from gemm_rocblas.cpp
void gemm_bf16_f32_rowmajor(const void*, const void*, float*, long long, long long,
                            long long, float, void*);
void gemm_f32_rowmajor(const float*, const float*, float*, long long, long long,
                       long long, float, void*);
void gemm_bf16_f32_rowmajor_lt(const void*, const void*, float*, long long, long long,
                               long long, float, void*);
void gemm_bf16_f32_nt_rowmajor_lt(const void*, const void*, float*, long long, long long,
                                  long long, float, void*);
void gemm_bf16_f32_nt_rowmajor(const void*, const void*, float*, long long, long long,
                               long long, float, void*);
}  // namespace raft_amd

namespace {

hipStream_t cur_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

void check_f32_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.dim() == 2, name, " must be 2D");
}

torch::Tensor reduce_rows(torch::Tensor x, int64_t op) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2);
  auto out = torch::empty({x.size(0)}, x.options());
  auto s = cur_stream();
  const long long m = x.size(0), d = x.size(1);
#define CASE(OPC)                                                                     \
  case OPC:                                                                           \
    if (x.scalar_type() == torch::kFloat32)                                           \
      raft_amd::launch_reduce_rows<OPC, float>(x.data_ptr<float>(), out.data_ptr<float>(), m, d, s); \
    else                                                                              \
      raft_amd::launch_reduce_rows<OPC, double>(x.data_ptr<double>(), out.data_ptr<double>(), m, d, s); \
    break;
  switch (op) { CASE(0) CASE(1) CASE(2) CASE(3) CASE(4) CASE(5)
    default: TORCH_CHECK(false, "bad op code"); }
#undef CASE
  return out;
}

torch::Tensor reduce_cols(torch::Tensor x, int64_t op) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2);
  auto out = torch::empty({x.size(1)}, x.options());
  auto s = cur_stream();
  const long long m = x.size(0), d = x.size(1);
#define CASE(OPC)                                                                     \
  case OPC:                                                                           \
    if (x.scalar_type() == torch::kFloat32)                                           \
      raft_amd::launch_reduce_cols<OPC, float>(x.data_ptr<float>(), out.data_ptr<float>(), m, d, s); \
    else                                                                              \
      raft_amd::launch_reduce_cols<OPC, double>(x.data_ptr<double>(), out.data_ptr<double>(), m, d, s); \
    break;
  switch (op) { CASE(0) CASE(1) CASE(2) CASE(3) CASE(4) CASE(5)
    default: TORCH_CHECK(false, "bad op code"); }
#undef CASE
  return out;
}

torch::Tensor rows_sqnorm_bf16(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.is_contiguous()
              && x.dim() == 2);
  auto out = torch::empty({x.size(0)}, x.options().dtype(torch::kFloat32));
  raft_amd::launch_rows_sqnorm_bf16(x.data_ptr(), out.data_ptr<float>(), x.size(0),
                                    x.size(1), cur_stream());
  return out;
}

torch::Tensor row_argmin(torch::Tensor x) {
  check_f32_2d(x, "x");
  auto out = torch::empty({x.size(0)}, x.options().dtype(torch::kInt32));
  raft_amd::launch_row_argmin(x.data_ptr<float>(), out.data_ptr<int>(), x.size(0),
                              x.size(1), cur_stream());
  return out;
}

torch::Tensor row_normalize_l2(torch::Tensor x, double eps) {
  check_f32_2d(x, "x");
  auto out = torch::empty_like(x);
  raft_amd::launch_row_normalize_l2(x.data_ptr<float>(), out.data_ptr<float>(),
                                    x.size(0), x.size(1), (float)eps, cur_stream());
  return out;
}

torch::Tensor l2_epilogue_(torch::Tensor g, torch::Tensor xn, torch::Tensor yn) {
  check_f32_2d(g, "g");
  raft_amd::launch_l2_epilogue(g.data_ptr<float>(), xn.data_ptr<float>(),
                               yn.data_ptr<float>(), g.size(0), g.size(1), cur_stream());
  return g;
}

void l2nn_epilogue(torch::Tensor g, torch::Tensor xn, torch::Tensor yn,
                   torch::Tensor dmin, torch::Tensor amin) {
  check_f32_2d(g, "g");
  raft_amd::launch_l2nn_epilogue(g.data_ptr<float>(), xn.data_ptr<float>(),
                                 yn.data_ptr<float>(), dmin.data_ptr<float>(),
                                 amin.data_ptr<int>(), g.size(0), g.size(1), cur_stream());
}

torch::Tensor pairwise_unexpanded(torch::Tensor x, torch::Tensor y, int64_t code,
                                  double p) {
  check_f32_2d(x, "x");
  check_f32_2d(y, "y");
  auto out = torch::empty({x.size(0), y.size(0)}, x.options());
  raft_amd::launch_pairwise_unexpanded(x.data_ptr<float>(), y.data_ptr<float>(),
                                       out.data_ptr<float>(), x.size(0), y.size(0),
                                       x.size(1), (int)code, (float)p, cur_stream());
  return out;
}

torch::Tensor rng_uniform(int64_t n, int64_t seed, int64_t subseq, int64_t device,
                          int64_t gen = 0) {
  auto out = torch::empty({n}, torch::dtype(torch::kFloat32).device(torch::kCUDA, device));
  raft_amd::launch_rng_uniform(out.data_ptr<float>(), n, (uint64_t)seed,
                               (uint64_t)subseq, cur_stream(), (int)gen);
  return out;
}

torch::Tensor rng_normal(int64_t n, int64_t seed, int64_t subseq, int64_t device,
                         int64_t gen = 0) {
  auto out = torch::empty({n}, torch::dtype(torch::kFloat32).device(torch::kCUDA, device));
  raft_amd::launch_rng_normal(out.data_ptr<float>(), n, (uint64_t)seed,
                              (uint64_t)subseq, cur_stream(), (int)gen);
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> make_blobs(int64_t n_rows, int64_t d,
                                                    torch::Tensor centers, double std_,
                                                    int64_t seed, int64_t subseq) {
  check_f32_2d(centers, "centers");
  auto x = torch::empty({n_rows, d}, centers.options());
  auto labels = torch::empty({n_rows}, centers.options().dtype(torch::kInt32));
  raft_amd::launch_make_blobs(x.data_ptr<float>(), labels.data_ptr<int>(),
                              centers.data_ptr<float>(), n_rows, d,
                              (int)centers.size(0), (float)std_, (uint64_t)seed,
                              (uint64_t)subseq, cur_stream());
  return {x, labels};
}

torch::Tensor csr_spmv(torch::Tensor indptr, torch::Tensor indices, torch::Tensor values,
                       torch::Tensor x, int64_t n_rows) {
  TORCH_CHECK(indptr.is_cuda() && indptr.scalar_type() == torch::kInt32);
  TORCH_CHECK(indices.is_cuda() && indices.scalar_type() == torch::kInt32);
  auto y = torch::empty({n_rows}, values.options());
  if (values.scalar_type() == torch::kFloat32) {
    raft_amd::launch_csr_spmv<float>(indptr.data_ptr<int>(), indices.data_ptr<int>(),
                                     values.data_ptr<float>(), x.data_ptr<float>(),
                                     y.data_ptr<float>(), n_rows, values.numel(),
                                     cur_stream());
  } else {
    raft_amd::launch_csr_spmv<double>(indptr.data_ptr<int>(), indices.data_ptr<int>(),
                                      values.data_ptr<double>(), x.data_ptr<double>(),
                                      y.data_ptr<double>(), n_rows, values.numel(),
                                      cur_stream());
  }
  return y;
}

torch::Tensor reduce_rows_by_key_sorted(torch::Tensor x, torch::Tensor perm,
                                        torch::Tensor keys_sorted, int64_t n_keys) {
  check_f32_2d(x, "x");
  TORCH_CHECK(perm.scalar_type() == torch::kInt32 && perm.is_contiguous());
  TORCH_CHECK(keys_sorted.scalar_type() == torch::kInt32 && keys_sorted.is_contiguous());
  auto sums = torch::zeros({n_keys, x.size(1)}, x.options());
  raft_amd::launch_reduce_rows_by_key_sorted(x.data_ptr<float>(), perm.data_ptr<int>(),
                                             keys_sorted.data_ptr<int>(),
                                             sums.data_ptr<float>(), nullptr,
                                             nullptr, nullptr,
                                             x.size(0), x.size(1), cur_stream());
  return sums;
}

void reduce_rows_by_key_sorted_into(torch::Tensor x, torch::Tensor perm,
                                    torch::Tensor keys_sorted, torch::Tensor sums,
                                    torch::Tensor counts,
                                    c10::optional<torch::Tensor> dmin,
                                    c10::optional<torch::Tensor> inertia_acc) {
  check_f32_2d(x, "x");
  TORCH_CHECK(perm.scalar_type() == torch::kInt32 && keys_sorted.scalar_type() == torch::kInt32);
  TORCH_CHECK(sums.is_contiguous() && counts.is_contiguous());
  raft_amd::launch_reduce_rows_by_key_sorted(
      x.data_ptr<float>(), perm.data_ptr<int>(), keys_sorted.data_ptr<int>(),
      sums.data_ptr<float>(), counts.data_ptr<float>(),
      dmin.has_value() ? dmin->data_ptr<float>() : nullptr,
      inertia_acc.has_value() ? inertia_acc->data_ptr<float>() : nullptr,
      x.size(0), x.size(1), cur_stream());
}

void split_bf16_norms(torch::Tensor c, std::vector<torch::Tensor> slices,
                      torch::Tensor cn,
                      c10::optional<torch::Tensor> cn_max = c10::nullopt) {
  check_f32_2d(c, "c");
  const int nslice = (int)slices.size();
  TORCH_CHECK(nslice >= 1 && nslice <= 3);
  for (auto& s : slices)
    TORCH_CHECK(s.scalar_type() == torch::kBFloat16 && s.is_contiguous()
                && s.sizes() == c.sizes());
  void* p0 = slices[0].data_ptr();
  void* p1 = nslice > 1 ? slices[1].data_ptr() : p0;
  void* p2 = nslice > 2 ? slices[2].data_ptr() : p0;
  raft_amd::launch_split_bf16_norms(
      c.data_ptr<float>(), p0, p1, p2, cn.data_ptr<float>(), nslice, c.size(0),
      c.size(1), cur_stream(),
      cn_max.has_value() ? cn_max->data_ptr<float>() : nullptr);
}

void kmeans_update_verify(torch::Tensor x, torch::Tensor perm,
                          torch::Tensor keys_sorted, torch::Tensor c,
                          torch::Tensor xn, torch::Tensor dmin, torch::Tensor amin,
                          torch::Tensor dmin2, torch::Tensor cn_max,
                          torch::Tensor sums, torch::Tensor counts,
                          c10::optional<torch::Tensor> inertia_acc,
                          double lead, double tail) {
  check_f32_2d(x, "x");
  check_f32_2d(c, "c");
  TORCH_CHECK(perm.scalar_type() == torch::kInt32 && keys_sorted.scalar_type() == torch::kInt32);
  TORCH_CHECK(sums.is_contiguous() && counts.is_contiguous());
  raft_amd::launch_kmeans_update_verify(
      x.data_ptr<float>(), perm.data_ptr<int>(), keys_sorted.data_ptr<int>(),
      c.data_ptr<float>(), xn.data_ptr<float>(), dmin.data_ptr<float>(),
      amin.data_ptr<int>(), dmin2.data_ptr<float>(), cn_max.data_ptr<float>(),
      sums.data_ptr<float>(), counts.data_ptr<float>(),
      inertia_acc.has_value() ? inertia_acc->data_ptr<float>() : nullptr,
      x.size(0), x.size(1), (int)c.size(0), cur_stream(), (float)lead,
      (float)tail);
}

void kmeans_update_centroids(torch::Tensor sums, torch::Tensor counts,
                             torch::Tensor centroids) {
  check_f32_2d(centroids, "centroids");
  raft_amd::launch_kmeans_update_centroids(sums.data_ptr<float>(),
                                           counts.data_ptr<float>(),
                                           centroids.data_ptr<float>(),
                                           centroids.size(0), centroids.size(1),
                                           cur_stream());
}

torch::Tensor reduce_rows_by_key(torch::Tensor x, torch::Tensor keys, int64_t n_keys) {
  check_f32_2d(x, "x");
  TORCH_CHECK(keys.scalar_type() == torch::kInt32);
  const long long kd = n_keys * x.size(1);
  // replica count: bound the workspace at ~32 MB, contention drop ~= replicas
  int replicas = (int)std::min<long long>(16, std::max<long long>(1, (32ll << 20) / (kd * 4)));
  auto work = torch::zeros({replicas, n_keys, x.size(1)}, x.options());
  auto sums = torch::empty({n_keys, x.size(1)}, x.options());
  raft_amd::launch_reduce_rows_by_key(x.data_ptr<float>(), keys.data_ptr<int>(),
                                      work.data_ptr<float>(), sums.data_ptr<float>(),
                                      x.size(0), x.size(1), n_keys, replicas,
                                      cur_stream());
  return sums;
}

std::tuple<torch::Tensor, torch::Tensor> select_k_generic(
    torch::Tensor x, c10::optional<torch::Tensor> row_off, int64_t k,
    bool select_min) {
  // x: [batch, len] dense, or flat [nnz] values with row_off [batch+1]
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  long long batch, stride, len_fixed;
  const long long* ro = nullptr;
  torch::Tensor ro_t;
  if (row_off.has_value()) {
    ro_t = row_off.value();
    TORCH_CHECK(ro_t.is_cuda() && ro_t.scalar_type() == torch::kInt64 &&
                ro_t.is_contiguous() && x.dim() == 1);
    batch = ro_t.numel() - 1;
    stride = 0;
    len_fixed = 0;
    ro = reinterpret_cast<const long long*>(ro_t.data_ptr<int64_t>());
  } else {
    TORCH_CHECK(x.dim() == 2);
    batch = x.size(0);
    stride = len_fixed = x.size(1);
  }
  auto out_v = torch::empty({batch, k}, x.options());
  auto out_i = torch::empty({batch, k}, x.options().dtype(torch::kInt64));
  auto* oi = reinterpret_cast<long long*>(out_i.data_ptr<int64_t>());
  switch (x.scalar_type()) {
    case torch::kFloat32:
      raft_amd::launch_select_k_generic_t<float>(
          x.data_ptr<float>(), ro, stride, len_fixed, out_v.data_ptr<float>(),
          oi, batch, k, select_min, cur_stream());
      break;
    case torch::kFloat64:
      raft_amd::launch_select_k_generic_t<double>(
          x.data_ptr<double>(), ro, stride, len_fixed,
          out_v.data_ptr<double>(), oi, batch, k, select_min, cur_stream());
      break;
    case torch::kBFloat16:
      raft_amd::launch_select_k_generic_t<__bf16>(
          reinterpret_cast<const __bf16*>(x.data_ptr<at::BFloat16>()), ro,
          stride, len_fixed, reinterpret_cast<__bf16*>(out_v.data_ptr<at::BFloat16>()),
          oi, batch, k, select_min, cur_stream());
      break;
    case torch::kHalf:
      raft_amd::launch_select_k_generic_t<_Float16>(
          reinterpret_cast<const _Float16*>(x.data_ptr<at::Half>()), ro,
          stride, len_fixed, reinterpret_cast<_Float16*>(out_v.data_ptr<at::Half>()),
          oi, batch, k, select_min, cur_stream());
      break;
    default:
      TORCH_CHECK(false, "select_k_generic: fp32/fp64/bf16/fp16 only");
  }
  return {out_v, out_i};
}

std::tuple<torch::Tensor, torch::Tensor> select_k(torch::Tensor x, int64_t k,
                                                  bool select_min, int64_t algo,
                                                  bool do_sort) {
  check_f32_2d(x, "x");
  auto vals = torch::empty({x.size(0), k}, x.options());
  auto idx = torch::empty({x.size(0), k}, x.options().dtype(torch::kInt32));
  // algo: 0 auto, 1 radix, 2 warpsort (k <= 64, wave-register queue)
  if (k <= 64 && algo == 2) {
    raft_amd::launch_select_k_warpsort(x.data_ptr<float>(), vals.data_ptr<float>(),
                                       idx.data_ptr<int>(), x.size(0), x.size(1),
                                       (int)k, select_min, cur_stream());
    return {vals, idx};
  }
  auto ws = torch::empty({raft_amd::select_k_workspace_bytes(x.size(0))},
                         x.options().dtype(torch::kUInt8));
  raft_amd::launch_select_k(x.data_ptr<float>(), vals.data_ptr<float>(),
                            idx.data_ptr<int>(), ws.data_ptr(), x.size(0), x.size(1),
                            (int)k, select_min, do_sort, cur_stream());
  return {vals, idx};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> fused_l2nn_split(
    std::vector<torch::Tensor> x_slices, std::vector<torch::Tensor> c_slices,
    torch::Tensor xn, torch::Tensor cn) {
  const int nslice = (int)x_slices.size();
  TORCH_CHECK(nslice >= 1 && nslice <= 3 && c_slices.size() == x_slices.size());
  const void* xsl[3];
  const void* csl[3];
  for (int s = 0; s < nslice; s++) {
    auto& xt = x_slices[s];
    auto& ct = c_slices[s];
    TORCH_CHECK(xt.is_cuda() && xt.scalar_type() == torch::kBFloat16 && xt.is_contiguous());
    TORCH_CHECK(ct.is_cuda() && ct.scalar_type() == torch::kBFloat16 && ct.is_contiguous());
    xsl[s] = xt.data_ptr();
    csl[s] = ct.data_ptr();
  }
  const long long m = x_slices[0].size(0);
  const long long n = c_slices[0].size(0);
  const long long d = x_slices[0].size(1);
  TORCH_CHECK(c_slices[0].size(1) == d, "dim mismatch");
  TORCH_CHECK(n % 128 == 0, "fused_l2nn: n must be a multiple of 128");
  TORCH_CHECK(d % 64 == 0, "fused_l2nn: d must be a multiple of 64");
  TORCH_CHECK(cn.numel() == n && xn.numel() == m);
  auto dmin = torch::empty({m}, xn.options());
  auto amin = torch::empty({m}, xn.options().dtype(torch::kInt32));
  auto dmin2 = torch::empty({m}, xn.options());
  static const bool use_persist = [] {
    const char* e = getenv("RAFT_AMD_PERSIST_L2NN");
    return e && e[0] == '1';
  }();  // measured slower than v1 at 10M x 256 (occupancy); kept for tuning
  static const bool no_w8 = [] {
    const char* e = getenv("RAFT_AMD_L2NN_W8");
    return e && e[0] == '0';
  }();
  if (!use_persist && raft_amd::fused_l2nn_256_supported(nslice, m, (int)n, (int)d)) {
    const long long n_col_tiles = n / 256;
    auto pd = torch::empty({n_col_tiles * m}, xn.options());
    auto pd2 = torch::empty({n_col_tiles * m}, xn.options());
    auto pi = torch::empty({n_col_tiles * m}, xn.options().dtype(torch::kInt32));
    raft_amd::launch_fused_l2nn_256(xsl, csl, xn.data_ptr<float>(),
                                    cn.data_ptr<float>(), pd.data_ptr<float>(),
                                    pd2.data_ptr<float>(), pi.data_ptr<int>(),
                                    dmin.data_ptr<float>(), amin.data_ptr<int>(),
                                    dmin2.data_ptr<float>(), m, (int)n, (int)d,
                                    nslice, cur_stream());
  } else if (!use_persist && raft_amd::fused_l2nn_2d_supported(nslice, m, (int)n, (int)d)) {
    const long long n_col_tiles = n / 128;
    auto pd = torch::empty({n_col_tiles * m}, xn.options());
    auto pd2 = torch::empty({n_col_tiles * m}, xn.options());
    auto pi = torch::empty({n_col_tiles * m}, xn.options().dtype(torch::kInt32));
    raft_amd::launch_fused_l2nn_2d(xsl, csl, xn.data_ptr<float>(),
                                   cn.data_ptr<float>(), pd.data_ptr<float>(),
                                   pd2.data_ptr<float>(), pi.data_ptr<int>(),
                                   dmin.data_ptr<float>(), amin.data_ptr<int>(),
                                   dmin2.data_ptr<float>(), m, (int)n, (int)d,
                                   nslice, cur_stream());
  } else if (!no_w8 && !use_persist && raft_amd::fused_l2nn_w8_supported(nslice, (int)n, (int)d)) {
    raft_amd::launch_fused_l2nn_w8(xsl, csl, xn.data_ptr<float>(),
                                   cn.data_ptr<float>(), dmin.data_ptr<float>(),
                                   amin.data_ptr<int>(), dmin2.data_ptr<float>(),
                                   m, (int)n, (int)d, nslice, cur_stream());
  } else if (use_persist && raft_amd::fused_l2nn_persist_supported(nslice, (int)d)) {
    raft_amd::launch_fused_l2nn_persist(xsl, csl, xn.data_ptr<float>(),
                                        cn.data_ptr<float>(), dmin.data_ptr<float>(),
                                        amin.data_ptr<int>(), dmin2.data_ptr<float>(),
                                        m, (int)n, (int)d, nslice, cur_stream());
  } else {
    raft_amd::launch_fused_l2nn_split(xsl, csl, xn.data_ptr<float>(),
                                      cn.data_ptr<float>(), dmin.data_ptr<float>(),
                                      amin.data_ptr<int>(), dmin2.data_ptr<float>(),
                                      m, (int)n, (int)d, nslice, cur_stream());
  }
  return {dmin, amin, dmin2};
}

torch::Tensor pairwise_l2_mfma(std::vector<torch::Tensor> x_slices,
                               std::vector<torch::Tensor> y_slices,
                               torch::Tensor xn, torch::Tensor yn,
                               c10::optional<torch::Tensor> out, bool sqrt_out) {
  const int nslice = (int)x_slices.size();
  TORCH_CHECK(nslice >= 1 && nslice <= 3 && y_slices.size() == x_slices.size());
  const void* xsl[3];
  const void* csl[3];
  for (int s = 0; s < nslice; s++) {
    TORCH_CHECK(x_slices[s].is_cuda() && x_slices[s].scalar_type() == torch::kBFloat16
                && x_slices[s].is_contiguous());
    TORCH_CHECK(y_slices[s].is_cuda() && y_slices[s].scalar_type() == torch::kBFloat16
                && y_slices[s].is_contiguous());
    xsl[s] = x_slices[s].data_ptr();
    csl[s] = y_slices[s].data_ptr();
  }
  const long long m = x_slices[0].size(0);
  const long long n = y_slices[0].size(0);
  const long long d = x_slices[0].size(1);
  TORCH_CHECK(d % 64 == 0, "pairwise_l2_mfma: d must be a multiple of 64");
  torch::Tensor o;
  if (out.has_value()) {
    o = out.value();
    TORCH_CHECK(o.is_contiguous() && o.size(0) >= m && o.size(1) == n);
  } else {
    o = torch::empty({m, n}, xn.options());
  }
  static const bool use_256 = [] {
    const char* e = getenv("RAFT_AMD_PW256");
    return e && e[0] == '1';
  }();
  if (use_256 && nslice <= 2 && m >= 512 && n >= 512) {
    // 256x256-tile kernel (BK=32 counted-vmcnt): 4x fewer workgroups, but
    // 1 block/CU serializes the tile-store phase with the K-loop — measured
    // 547 vs the 128^2 kernel's 736 Gdist/s at 1Mx128 (round 2); kept for
    // A/B via RAFT_AMD_PW256=1
    raft_amd::launch_pairwise_l2_mfma256(xsl, csl, xn.data_ptr<float>(),
                                         yn.data_ptr<float>(), o.data_ptr<float>(),
                                         m, n, (int)d, o.size(1), nslice, sqrt_out,
                                         cur_stream());
  } else {
    raft_amd::launch_pairwise_l2_mfma(xsl, csl, xn.data_ptr<float>(), yn.data_ptr<float>(),
                                      o.data_ptr<float>(), m, n, (int)d, o.size(1),
                                      nslice, sqrt_out, cur_stream());
  }
  return o;
}

void l2nn_verify_repair(torch::Tensor x, torch::Tensor c, torch::Tensor xn,
                        torch::Tensor dmin, torch::Tensor amin, torch::Tensor dmin2,
                        torch::Tensor cn_max, double lead, double tail) {
  check_f32_2d(x, "x");
  check_f32_2d(c, "c");
  TORCH_CHECK(cn_max.scalar_type() == torch::kFloat32 && cn_max.numel() >= 1);
  raft_amd::launch_l2nn_verify_repair(x.data_ptr<float>(), c.data_ptr<float>(),
                                      xn.data_ptr<float>(), dmin.data_ptr<float>(),
                                      amin.data_ptr<int>(), dmin2.data_ptr<float>(),
                                      cn_max.data_ptr<float>(), x.size(0),
                                      (int)c.size(0), (int)x.size(1), cur_stream(),
                                      (float)lead, (float)tail);
}

void pairwise_l2_filter(std::vector<torch::Tensor> x_slices,
                        std::vector<torch::Tensor> y_slices, torch::Tensor xn,
                        torch::Tensor yn, torch::Tensor thr, torch::Tensor out_d,
                        torch::Tensor out_i, torch::Tensor cnt, int64_t col_offset) {
  const int nslice = (int)x_slices.size();
  TORCH_CHECK(nslice >= 1 && nslice <= 3 && y_slices.size() == x_slices.size());
  const void* xsl[3];
  const void* csl[3];
  for (int s = 0; s < nslice; s++) {
    TORCH_CHECK(x_slices[s].is_cuda() && x_slices[s].scalar_type() == torch::kBFloat16
                && x_slices[s].is_contiguous());
    TORCH_CHECK(y_slices[s].is_cuda() && y_slices[s].scalar_type() == torch::kBFloat16
                && y_slices[s].is_contiguous());
    xsl[s] = x_slices[s].data_ptr();
    csl[s] = y_slices[s].data_ptr();
  }
  const long long m = x_slices[0].size(0);
  const long long n = y_slices[0].size(0);
  const long long d = x_slices[0].size(1);
  TORCH_CHECK(d % 64 == 0, "pairwise_l2_filter: d must be a multiple of 64");
  const int cap = (int)out_d.size(1);
  TORCH_CHECK(out_d.size(0) == m && out_i.sizes() == out_d.sizes());
  TORCH_CHECK(cnt.scalar_type() == torch::kInt32 && cnt.numel() == m);
  static const bool use_f256 = [] {
    const char* e = getenv("RAFT_AMD_FILTER256");
    return e && e[0] == '1';
  }();  // A/B @100M-row kNN: 128-tile 9006 q/s vs 256-tile 7226 — the 4-wave
        // 2-block filter wins (same pattern as the fused L2-NN A/B)
  if (use_f256 && nslice <= 2 && m >= 512 && n >= 512) {
    raft_amd::launch_pairwise_l2_filter256(xsl, csl, xn.data_ptr<float>(),
                                           yn.data_ptr<float>(), thr.data_ptr<float>(),
                                           out_d.data_ptr<float>(), out_i.data_ptr<int>(),
                                           cnt.data_ptr<int>(), cap, col_offset, m, n,
                                           (int)d, nslice, cur_stream());
  } else {
    raft_amd::launch_pairwise_l2_filter(xsl, csl, xn.data_ptr<float>(),
                                        yn.data_ptr<float>(), thr.data_ptr<float>(),
                                        out_d.data_ptr<float>(), out_i.data_ptr<int>(),
                                        cnt.data_ptr<int>(), cap, col_offset, m, n,
                                        (int)d, nslice, cur_stream());
  }
}

// backend toggle (reference parity: cublas vs cublasLt wrapper pair) —
// RAFT_AMD_GEMM_BACKEND=hipblaslt routes the bf16->f32 GEMMs through the
// hipBLASLt heuristic path (csrc/gemm_hipblaslt.cpp)
static bool use_hipblaslt() {
  static const bool on = [] {
    const char* e = getenv("RAFT_AMD_GEMM_BACKEND");
    return e && std::string(e) == "hipblaslt";
  }();
  return on;
}

torch::Tensor gemm_bf16_f32(torch::Tensor a, torch::Tensor b,
                            c10::optional<torch::Tensor> out, double beta) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(b.is_cuda() && b.scalar_type() == torch::kBFloat16 && b.is_contiguous());
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(0));
  torch::Tensor c;
  float bt = (float)beta;
  if (out.has_value()) {
    c = out.value();
    TORCH_CHECK(c.scalar_type() == torch::kFloat32 && c.is_contiguous());
  } else {
    c = torch::empty({a.size(0), b.size(1)},
                     a.options().dtype(torch::kFloat32));
    bt = 0.0f;
  }
  if (use_hipblaslt())
    raft_amd::gemm_bf16_f32_rowmajor_lt(a.data_ptr(), b.data_ptr(), c.data_ptr<float>(),
                                        a.size(0), b.size(1), a.size(1), bt,
                                        (void*)cur_stream());
  else
    raft_amd::gemm_bf16_f32_rowmajor(a.data_ptr(), b.data_ptr(), c.data_ptr<float>(),
                                     a.size(0), b.size(1), a.size(1), bt,
                                     (void*)cur_stream());
  return c;
}

torch::Tensor gemm_bf16_f32_nt(torch::Tensor a, torch::Tensor b,
                               c10::optional<torch::Tensor> out, double beta) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(b.is_cuda() && b.scalar_type() == torch::kBFloat16 && b.is_contiguous());
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1));
  torch::Tensor c;
  float bt = (float)beta;
  if (out.has_value()) {
    c = out.value();
    TORCH_CHECK(c.scalar_type() == torch::kFloat32 && c.is_contiguous());
  } else {
    c = torch::empty({a.size(0), b.size(0)}, a.options().dtype(torch::kFloat32));
    bt = 0.0f;
  }
  if (use_hipblaslt())
    raft_amd::gemm_bf16_f32_nt_rowmajor_lt(a.data_ptr(), b.data_ptr(), c.data_ptr<float>(),
                                           a.size(0), b.size(0), a.size(1), bt,
                                           (void*)cur_stream());
  else
    raft_amd::gemm_bf16_f32_nt_rowmajor(a.data_ptr(), b.data_ptr(), c.data_ptr<float>(),
                                        a.size(0), b.size(0), a.size(1), bt,
                                        (void*)cur_stream());
  return c;
}

torch::Tensor gemm_f32(torch::Tensor a, torch::Tensor b) {
  check_f32_2d(a, "a");
  check_f32_2d(b, "b");
  auto c = torch::empty({a.size(0), b.size(1)}, a.options());
  raft_amd::gemm_f32_rowmajor(a.data_ptr<float>(), b.data_ptr<float>(),
                              c.data_ptr<float>(), a.size(0), b.size(1), a.size(1),
                              0.0f, (void*)cur_stream());
  return c;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("reduce_rows", &reduce_rows, "rowwise reduction (op code)");
  m.def("lanczos_pre_", [](torch::Tensor u, torch::Tensor v_i,
                           c10::optional<torch::Tensor> v_prev,
                           c10::optional<torch::Tensor> beta,
                           torch::Tensor alpha_out) {
    raft_amd::launch_lanczos_pre(
        u.data_ptr<float>(), v_i.data_ptr<float>(),
        v_prev.has_value() ? v_prev->data_ptr<float>() : nullptr,
        beta.has_value() ? beta->data_ptr<float>() : nullptr,
        alpha_out.data_ptr<double>(), u.numel(), cur_stream());
  }, "fused: u -= beta*v_prev; alpha_out = dot(v_i, u)");
  m.def("lanczos_sub_alpha_", [](torch::Tensor u, torch::Tensor v_i,
                                 torch::Tensor alpha, torch::Tensor t_diag) {
    raft_amd::launch_lanczos_sub_alpha(u.data_ptr<float>(),
                                       v_i.data_ptr<float>(),
                                       alpha.data_ptr<double>(),
                                       t_diag.data_ptr<float>(), u.numel(),
                                       cur_stream());
  }, "fused: u -= alpha*v_i; t_diag[0] = alpha");
  m.def("lanczos_norm2_", [](torch::Tensor u, torch::Tensor norm2_out) {
    raft_amd::launch_lanczos_norm2(u.data_ptr<float>(),
                                   norm2_out.data_ptr<double>(), u.numel(),
                                   cur_stream());
  }, "norm2_out[0] = ||u||^2 (fp64 accumulate)");
  m.def("lanczos_normalize_", [](torch::Tensor u, torch::Tensor v_next,
                                 torch::Tensor norm2,
                                 c10::optional<torch::Tensor> t_up,
                                 c10::optional<torch::Tensor> t_dn,
                                 c10::optional<torch::Tensor> beta_out) {
    raft_amd::launch_lanczos_normalize(
        u.data_ptr<float>(), v_next.data_ptr<float>(),
        norm2.data_ptr<double>(),
        t_up.has_value() ? t_up->data_ptr<float>() : nullptr,
        t_dn.has_value() ? t_dn->data_ptr<float>() : nullptr,
        beta_out.has_value() ? beta_out->data_ptr<float>() : nullptr,
        u.numel(), cur_stream());
  }, "fused: v_next = u/||u||; t couplings = ||u||");
  m.def("lanczos_cycle_", [](torch::Tensor indptr, torch::Tensor indices,
                             torch::Tensor values, torch::Tensor v,
                             torch::Tensor t_mat, torch::Tensor u,
                             torch::Tensor w, torch::Tensor v_next,
                             torch::Tensor beta_out, torch::Tensor alpha_scal,
                             torch::Tensor norm_scal, int64_t start,
                             int64_t ncv) {
    // Whole ncv-step extension cycle driven from C++: the per-step python
    // dispatch (~1 ms/step measured at 10M rows) disappears — one host call
    // per restart cycle. Mirrors sparse/solver/lanczos.py _extend exactly:
    // SpMV, (arrowhead | beta-recurrence) + alpha dot, alpha subtract,
    // 2-pass CGS reorth (rocBLAS sgemv pairs), norm, normalize.
    TORCH_CHECK(v.is_cuda() && v.is_contiguous() &&
                v.scalar_type() == torch::kFloat32);
    TORCH_CHECK(t_mat.is_contiguous() && t_mat.size(0) == ncv &&
                t_mat.size(1) == ncv);
    TORCH_CHECK(indptr.scalar_type() == torch::kInt32 &&
                indices.scalar_type() == torch::kInt32 &&
                values.scalar_type() == torch::kFloat32);
    TORCH_CHECK(u.numel() == v.size(1) && v_next.numel() == v.size(1) &&
                w.numel() >= ncv);
    const long long n = v.size(1);
    const long long nnz = indices.numel();
    float* vp = v.data_ptr<float>();
    float* tp = t_mat.data_ptr<float>();
    float* up = u.data_ptr<float>();
    float* wp = w.data_ptr<float>();
    double* ap = alpha_scal.data_ptr<double>();
    double* np2 = norm_scal.data_ptr<double>();
    hipStream_t s = cur_stream();
    for (long long i = start; i < ncv; i++) {
      raft_amd::launch_csr_spmv<float>(
          indptr.data_ptr<int>(), indices.data_ptr<int>(),
          values.data_ptr<float>(), vp + i * n, up, n, nnz, s);
      if (i == start && start > 0) {
        // arrowhead couplings: u -= V[:start]^T t_mat[start, :start]
        raft_amd::sgemv_rowmajor(vp, tp + start * ncv, up, start, n,
                                 /*trans=*/true, -1.f, 1.f, s);
        raft_amd::launch_lanczos_pre(up, vp + i * n, nullptr, nullptr, ap, n, s);
      } else if (i > start) {
        raft_amd::launch_lanczos_pre(up, vp + i * n, vp + (i - 1) * n,
                                     tp + i * ncv + (i - 1), ap, n, s);
      } else {
        raft_amd::launch_lanczos_pre(up, vp + i * n, nullptr, nullptr, ap, n, s);
      }
      raft_amd::launch_lanczos_sub_alpha(up, vp + i * n, ap, tp + i * ncv + i,
                                         n, s);
      for (int pass = 0; pass < 2; pass++) {
        raft_amd::sgemv_rowmajor(vp, up, wp, i + 1, n, false, 1.f, 0.f, s);
        raft_amd::sgemv_rowmajor(vp, wp, up, i + 1, n, true, -1.f, 1.f, s);
      }
      raft_amd::launch_lanczos_norm2(up, np2, n, s);
      if (i + 1 < ncv) {
        raft_amd::launch_lanczos_normalize(up, vp + (i + 1) * n, np2,
                                           tp + i * ncv + (i + 1),
                                           tp + (i + 1) * ncv + i, nullptr, n,
                                           s);
      } else {
        raft_amd::launch_lanczos_normalize(up, v_next.data_ptr<float>(), np2,
                                           nullptr, nullptr,
                                           beta_out.data_ptr<float>(), n, s);
      }
    }
  }, "full Lanczos extension cycle (SpMV + CGS reorth + fused steps) in C++");
  m.def("cholesky_r1_update_", [](torch::Tensor l, torch::Tensor x) {
    TORCH_CHECK(l.is_cuda() && l.dim() == 2 && l.size(0) == l.size(1) &&
                l.is_contiguous() && x.is_contiguous() &&
                x.numel() == l.size(0));
    const int n = (int)l.size(0);
    if (l.scalar_type() == torch::kFloat32) {
      raft_amd::launch_cholesky_r1_update_f32(l.data_ptr<float>(),
                                              x.data_ptr<float>(), n, n,
                                              cur_stream());
    } else if (l.scalar_type() == torch::kFloat64) {
      raft_amd::launch_cholesky_r1_update_f64(l.data_ptr<double>(),
                                              x.data_ptr<double>(), n, n,
                                              cur_stream());
    } else {
      TORCH_CHECK(false, "cholesky_r1_update_: fp32/fp64 only");
    }
  }, "in-place rank-1 Cholesky update (single-kernel hyperbolic rotations)");
  m.def("reduce_cols", &reduce_cols, "columnwise reduction (op code)");
  m.def("row_argmin", &row_argmin, "rowwise argmin");
  m.def("rows_sqnorm_bf16", &rows_sqnorm_bf16, "bf16 row squared norms -> f32");
  m.def("row_normalize_l2", &row_normalize_l2, "fused L2 row normalize");
  m.def("l2_epilogue_", &l2_epilogue_, "in-place L2 distance epilogue");
  m.def("l2nn_epilogue", &l2nn_epilogue, "fused argmin epilogue over GEMM tile");
  m.def("pairwise_unexpanded", &pairwise_unexpanded, "tiled unexpanded distances");
  m.def("rng_uniform", &rng_uniform, "counter-based uniform [0,1)",
        pybind11::arg("n"), pybind11::arg("seed"), pybind11::arg("subseq"),
        pybind11::arg("device"), pybind11::arg("gen") = 0);
  m.def("rng_normal", &rng_normal, "counter-based Box-Muller standard normal",
        pybind11::arg("n"), pybind11::arg("seed"), pybind11::arg("subseq"),
        pybind11::arg("device"), pybind11::arg("gen") = 0);
  m.def("make_blobs", &make_blobs, "fused gaussian blob generator");
  m.def("csr_spmv", &csr_spmv, "CSR SpMV (sub-wave per row)");
  m.def("reduce_rows_by_key", &reduce_rows_by_key, "keyed row accumulation");
  m.def("reduce_rows_by_key_sorted", &reduce_rows_by_key_sorted,
        "keyed row accumulation over a key-sorted permutation");
  m.def("reduce_rows_by_key_sorted_into", &reduce_rows_by_key_sorted_into,
        pybind11::arg("x"), pybind11::arg("perm"), pybind11::arg("keys_sorted"),
        pybind11::arg("sums"), pybind11::arg("counts"),
        pybind11::arg("dmin") = pybind11::none(),
        pybind11::arg("inertia_acc") = pybind11::none(),
        "keyed row accumulation + counts into caller buffers");
  m.def("split_bf16_norms", &split_bf16_norms,
        pybind11::arg("c"), pybind11::arg("slices"), pybind11::arg("cn"),
        pybind11::arg("cn_max") = pybind11::none(),
        "fused fp32->bf16 slice split + row sq-norms");
  m.def("kmeans_update_verify", &kmeans_update_verify,
        pybind11::arg("x"), pybind11::arg("perm"), pybind11::arg("keys_sorted"),
        pybind11::arg("c"), pybind11::arg("xn"), pybind11::arg("dmin"),
        pybind11::arg("amin"), pybind11::arg("dmin2"), pybind11::arg("cn_max"),
        pybind11::arg("sums"), pybind11::arg("counts"),
        pybind11::arg("inertia_acc") = pybind11::none(),
        pybind11::arg("lead") = 0x1p-13, pybind11::arg("tail") = 0x1p-18,
        "fused centroid-sum accumulation + exact-fp32 verify/refine (one X pass)");
  m.def("kmeans_update_centroids", &kmeans_update_centroids,
        "centroids = counts>0 ? sums/counts : centroids");
  m.def("select_k", &select_k, "batched top-k (radix)");
  m.def("device_sample_test", [](torch::Tensor weights, int64_t n_draws,
                                 int64_t seed) {
    TORCH_CHECK(weights.is_cuda() && weights.numel() == 256 &&
                weights.scalar_type() == torch::kFloat32);
    auto out = torch::empty({n_draws}, weights.options().dtype(torch::kInt32));
    raft_amd::launch_device_sample_test(weights.contiguous().data_ptr<float>(),
                                        out.data_ptr<int>(), (int)n_draws,
                                        (uint64_t)seed, cur_stream());
    return out;
  }, "block_random_sample test driver (weighted in-kernel selection)");
  m.def("sddmm", [](torch::Tensor a, torch::Tensor b, torch::Tensor rows,
                    torch::Tensor cols) {
    TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous() &&
                a.scalar_type() == torch::kFloat32 &&
                a.size(1) == b.size(1) &&
                rows.scalar_type() == torch::kInt32 &&
                cols.scalar_type() == torch::kInt32);
    auto vals = torch::empty({rows.numel()}, a.options());
    raft_amd::launch_sddmm(a.data_ptr<float>(), b.data_ptr<float>(),
                           rows.contiguous().data_ptr<int>(),
                           cols.contiguous().data_ptr<int>(),
                           vals.data_ptr<float>(), rows.numel(), a.size(1),
                           cur_stream());
    return vals;
  }, "sampled dense-dense matmul: vals[e] = dot(a[rows[e]], b[cols[e]])");
  m.def("linewise", [](torch::Tensor x, torch::Tensor v1,
                       c10::optional<torch::Tensor> v2, bool along_rows,
                       int64_t op1, int64_t op2) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&
                x.scalar_type() == torch::kFloat32);
    const long long n = x.size(0), d = x.size(1);
    TORCH_CHECK(v1.numel() == (along_rows ? d : n) && v1.is_contiguous());
    const float* v2p = nullptr;
    torch::Tensor v2t;
    if (v2.has_value()) {
      v2t = v2.value().contiguous();
      TORCH_CHECK(v2t.numel() == v1.numel());
      v2p = v2t.data_ptr<float>();
    }
    auto out = torch::empty_like(x);
    raft_amd::launch_linewise(x.data_ptr<float>(), out.data_ptr<float>(),
                              v1.data_ptr<float>(), v2p, n, d, along_rows,
                              (int)op1, (int)op2, cur_stream());
    return out;
  }, "fused linewise broadcast: out = (x op1 v1) [op2 v2]",
        pybind11::arg("x"), pybind11::arg("v1"),
        pybind11::arg("v2") = pybind11::none(),
        pybind11::arg("along_rows") = true, pybind11::arg("op1") = 0,
        pybind11::arg("op2") = 0);
  m.def("histogram_f32", [](torch::Tensor x, int64_t n_bins, double lo, double hi) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&
                x.scalar_type() == torch::kFloat32);
    auto out = torch::zeros({n_bins, x.size(1)},
                            x.options().dtype(torch::kInt64));
    raft_amd::launch_histogram(
        x.data_ptr<float>(), x.size(0), x.size(1), (int)n_bins, (float)lo,
        (float)hi,
        reinterpret_cast<unsigned long long*>(out.data_ptr<int64_t>()),
        cur_stream());
    return out;
  }, "per-column histograms (LDS-multi/LDS/gmem strategies by n_bins)");
  m.def("bitset_set_", [](torch::Tensor words, torch::Tensor idx, bool value) {
    TORCH_CHECK(words.is_cuda() && words.scalar_type() == torch::kInt32 &&
                words.is_contiguous() && idx.scalar_type() == torch::kInt64);
    raft_amd::launch_bitset_set(
        reinterpret_cast<unsigned int*>(words.data_ptr<int>()),
        reinterpret_cast<const long long*>(idx.contiguous().data_ptr<int64_t>()),
        idx.numel(), value ? 1 : 0, cur_stream());
  }, "O(k) bitset scatter set/clear");
  m.def("bitset_test", [](torch::Tensor words, torch::Tensor idx) {
    TORCH_CHECK(words.is_cuda() && words.scalar_type() == torch::kInt32);
    auto out = torch::empty({idx.numel()}, words.options().dtype(torch::kBool));
    raft_amd::launch_bitset_test(
        reinterpret_cast<const unsigned int*>(words.data_ptr<int>()),
        reinterpret_cast<const long long*>(idx.contiguous().data_ptr<int64_t>()),
        out.data_ptr<bool>(), idx.numel(), cur_stream());
    return out;
  }, "bitset membership test");
  m.def("bitset_count", [](torch::Tensor words) {
    TORCH_CHECK(words.is_cuda() && words.scalar_type() == torch::kInt32);
    auto out = torch::zeros({1}, words.options().dtype(torch::kInt64));
    raft_amd::launch_bitset_count(
        reinterpret_cast<const unsigned int*>(words.data_ptr<int>()),
        words.numel(),
        reinterpret_cast<unsigned long long*>(out.data_ptr<int64_t>()),
        cur_stream());
    return out;
  }, "bitset popcount");
  m.def("select_k_generic", &select_k_generic,
        "generic top-k: any dtype, unbounded k, int64 idx, CSR row offsets",
        pybind11::arg("x"), pybind11::arg("row_off") = pybind11::none(),
        pybind11::arg("k"), pybind11::arg("select_min") = true);
  m.def("pairwise_l2_filter", &pairwise_l2_filter,
        "threshold-filtered pairwise L2 candidate emission (fused kNN)");
  m.def("pairwise_l2_mfma", &pairwise_l2_mfma,
        "fused split-bf16 MFMA pairwise L2 tile (single-write epilogue)",
        pybind11::arg("x_slices"), pybind11::arg("y_slices"), pybind11::arg("xn"),
        pybind11::arg("yn"), pybind11::arg("out") = pybind11::none(),
        pybind11::arg("sqrt_out") = false);
  m.def("l2nn_verify_repair", &l2nn_verify_repair,
        pybind11::arg("x"), pybind11::arg("c"), pybind11::arg("xn"),
        pybind11::arg("dmin"), pybind11::arg("amin"), pybind11::arg("dmin2"),
        pybind11::arg("cn_max"),
        pybind11::arg("lead") = 0x1p-13, pybind11::arg("tail") = 0x1p-18,
        "exact-fp32 verification/repair of split-bf16 L2-NN results");
  m.def("fused_l2nn_split", &fused_l2nn_split,
        "fused split-bf16 MFMA L2-NN (distance + argmin, no materialization)");
  m.def("gemm_bf16_f32", &gemm_bf16_f32, "bf16 x bf16 -> f32 rocBLAS gemm_ex",
        pybind11::arg("a"), pybind11::arg("b"), pybind11::arg("out") = pybind11::none(),
        pybind11::arg("beta") = 0.0);
  m.def("gemm_bf16_f32_nt", &gemm_bf16_f32_nt,
        "bf16 A @ B^T -> f32 (both row-major) rocBLAS gemm_ex",
        pybind11::arg("a"), pybind11::arg("b"), pybind11::arg("out") = pybind11::none(),
        pybind11::arg("beta") = 0.0);
  m.def("gemm_f32", &gemm_f32, "fp32 rocBLAS sgemm");
}
