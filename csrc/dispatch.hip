// dispatch.hip — host-side launch layer for the CDNA4 SGEMM kernel family.
// Reference-parity role: the if/else kernel dispatch of
// /root/reference/kernel/ft_sgemm/sgemm.cu:110-199, driven by the generated
// tier table instead of hand-unrolled launches.

#include <hip/hip_runtime.h>

#include "ft_core.h"
#include "ft_kernels.hpp"
#include "generated/tile_params.h"

namespace ftsgemm {

template <int BM, int BN, int BK, int WM, int WN, int MM>
static hipError_t launch_tier(bool abft, bool inject, int M, int N, int K,
                              const float* A, const float* B, float* C,
                              float alpha, float beta, float tau,
                              float inj_mag, hipStream_t stream) {
  if (M % BM || N % BN || K % BK || M % 4 || N % 4)
    return hipErrorInvalidValue;
  dim3 grid(M / BM, N / BN);
  dim3 block(64 * (BM / WM) * (BN / WN));
  const int niter = K / BK;
  // ~20 verify/inject windows per GEMM (reference period K/20,
  // ft_sgemm_huge.cuh:324-327), rounded to whole BK panels.
  int stride = niter / 20;
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

hipError_t sgemm_tier_launch(int tier, bool abft, bool inject, int M, int N,
                             int K, const float* A, const float* B, float* C,
                             float alpha, float beta, float tau,
                             float inj_mag, hipStream_t stream) {
  switch (tier) {
#define FT_CASE(name, BM, BN, BK, WM, WN, MM)                              \
  case FT_TIER_ID_##name:                                                  \
    return launch_tier<BM, BN, BK, WM, WN, MM>(abft, inject, M, N, K, A,   \
                                               B, C, alpha, beta, tau,     \
                                               inj_mag, stream);
    FT_TIER_LIST(FT_CASE)
#undef FT_CASE
    default:
      return hipErrorInvalidValue;
  }
}

bool sgemm_tier_supported(int tier, int M, int N, int K) {
  switch (tier) {
#define FT_CHK(name, BM, BN, BK, WM, WN, MM)                               \
  case FT_TIER_ID_##name:                                                  \
    return !(M % BM || N % BN || K % BK || M % 4 || N % 4);
    FT_TIER_LIST(FT_CHK)
#undef FT_CHK
    default:
      return false;
  }
}

}  // namespace ftsgemm
