// tier_launch.hpp — shared launcher template; each tier is compiled in its
// own translation unit (csrc/generated/kernel_<tier>.hip) so the register
// allocator sees one tier's three variants at a time (co-compiled template
// instantiations perturb each other's codegen — measured 88 vs 112 TF on
// the fused-ABFT huge tier in a monolithic TU).
#pragma once

#include <hip/hip_runtime.h>

#include "ft_kernels.hpp"

namespace ftsgemm {

template <int BM, int BN, int BK, int WM, int WN, int MM>
hipError_t launch_tier(bool abft, bool inject, int M, int N, int K,
                       const float* A, const float* B, float* C, float alpha,
                       float beta, float tau, float inj_mag,
                       int verify_windows, hipStream_t stream) {
  if (M % BM || N % BN || K % BK || M % 4 || N % 4) return hipErrorInvalidValue;
  dim3 grid(M / BM, N / BN);
  dim3 block(64 * (BM / WM) * (BN / WN));
  const int niter = K / BK;
  // ~20 verify/inject windows per GEMM (reference period K/20,
  // ft_sgemm_huge.cuh:324-327), whole BK panels; plain kernels run one
  // burst.
  int stride = abft ? niter / (verify_windows > 0 ? verify_windows : 20)
                    : niter;
  if (stride < 1) stride = 1;
  if (abft && inject) {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BK, WM, WN, MM, true, true>), grid,
                       block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag);
  } else if (abft) {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BK, WM, WN, MM, true, false>),
                       grid, block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag);
  } else {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BK, WM, WN, MM, false, false>),
                       grid, block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag);
  }
  return hipGetLastError();
}

}  // namespace ftsgemm
