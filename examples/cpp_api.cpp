// C++ API example: link against the kernel TUs + dispatch + rocblas_path
// (see Makefile's `cli` target for the exact compile line).
//
//   hipcc -x hip --offload-arch=gfx950 -O3 -std=c++17 -Icsrc \
//     examples/cpp_api.cpp csrc/dispatch.hip csrc/rocblas_path.hip \
//     csrc/generated/kernel_*.hip -L/opt/rocm/lib -lrocblas -o example
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

#include "ft_core.h"
#include "generated/tile_params.h"

int main() {
  const int n = 2048;
  const size_t bytes = (size_t)n * n * sizeof(float);
  float *dA, *dB, *dC, *ws;
  hipMalloc(&dA, bytes);
  hipMalloc(&dB, bytes);
  hipMalloc(&dC, bytes);
  std::vector<float> h(n * (size_t)n, 0.5f);
  hipMemcpy(dA, h.data(), bytes, hipMemcpyHostToDevice);
  hipMemcpy(dB, h.data(), bytes, hipMemcpyHostToDevice);

  // fused-ABFT huge tier with the 20-fault self-test
  size_t wsf = ftsgemm::sgemm_abft_workspace_floats(FT_TIER_ID_huge, n, n, n);
  hipMalloc(&ws, wsf * sizeof(float));
  hipError_t err = ftsgemm::sgemm_tier_launch(
      FT_TIER_ID_huge, /*abft=*/true, /*inject=*/true, n, n, n, dA, dB, dC,
      /*alpha=*/1.f, /*beta=*/0.f, /*tau=*/9500.f, /*inj_mag=*/10000.f,
      /*verify_windows=*/20, ws, /*stream=*/0);
  if (err != hipSuccess) {
    fprintf(stderr, "launch failed: %s\n", hipGetErrorString(err));
    return 1;
  }
  hipDeviceSynchronize();
  printf("fused-ABFT GEMM done\n");
  return 0;
}
