// C++ API smoke test: exercises the native raft_amd kernels through the
// public header (include/raft_amd/raft_amd.hpp) with NO Python involved —
// the raft_runtime-style consumability check (compiled+run by
// tests/test_cpp_api.py on a GPU box against the build/ext objects).
#include <raft_amd/raft_amd.hpp>

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <vector>

#define CHECK(c)                                                   \
  if (!(c)) {                                                      \
    std::fprintf(stderr, "FAILED: %s (line %d)\n", #c, __LINE__);  \
    return 1;                                                      \
  }

#define HIP_OK(e) CHECK((e) == hipSuccess)

int main() {
  const long long n_rows = 1024, d = 256;

  // RNG: uniform draws
  float* du;
  HIP_OK(hipMalloc(&du, n_rows * d * sizeof(float)));
  raft_amd::launch_rng_uniform(du, n_rows * d, 42, 0, nullptr);
  HIP_OK(hipDeviceSynchronize());
  std::vector<float> hu(n_rows * d);
  HIP_OK(hipMemcpy(hu.data(), du, hu.size() * 4, hipMemcpyDeviceToHost));
  double mean = 0;
  for (float v : hu) {
    CHECK(v >= 0.f && v < 1.f);
    mean += v;
  }
  mean /= hu.size();
  CHECK(std::fabs(mean - 0.5) < 0.01);

  // row reduction (sum) vs host
  float* dout;
  HIP_OK(hipMalloc(&dout, n_rows * sizeof(float)));
  raft_amd::launch_reduce_rows<0, float>(du, dout, n_rows, d, nullptr);
  HIP_OK(hipDeviceSynchronize());
  std::vector<float> hout(n_rows);
  HIP_OK(hipMemcpy(hout.data(), dout, n_rows * 4, hipMemcpyDeviceToHost));
  for (int r = 0; r < 8; r++) {
    double ref = 0;
    for (int j = 0; j < d; j++) ref += hu[r * d + j];
    CHECK(std::fabs(hout[r] - ref) < 1e-3 * d);
  }

  // argmin
  int* dmin_i;
  HIP_OK(hipMalloc(&dmin_i, n_rows * sizeof(int)));
  raft_amd::launch_row_argmin(du, dmin_i, n_rows, d, nullptr);
  HIP_OK(hipDeviceSynchronize());
  std::vector<int> hmin(n_rows);
  HIP_OK(hipMemcpy(hmin.data(), dmin_i, n_rows * 4, hipMemcpyDeviceToHost));
  for (int r = 0; r < 8; r++) {
    int ref = 0;
    for (int j = 1; j < d; j++)
      if (hu[r * d + j] < hu[r * d + ref]) ref = j;
    CHECK(hmin[r] == ref);
  }

  std::printf("cpp smoke OK (mean=%.4f)\n", mean);
  HIP_OK(hipFree(du));
  HIP_OK(hipFree(dout));
  HIP_OK(hipFree(dmin_i));
  return 0;
}
