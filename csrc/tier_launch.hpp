// tier_launch.hpp — shared launcher template; each tier is compiled in its
// own translation unit (csrc/generated/kernel_<tier>.hip) so the register
// allocator sees one tier's three variants at a time (co-compiled template
// instantiations perturb each other's codegen — measured 88 vs 112 TF on
// the fused-ABFT huge tier in a monolithic TU).
#pragma once

#include <hip/hip_runtime.h>

#include "ft_kernels.hpp"

namespace ftsgemm {

// ABFT workspace layout: [ (M/WM) rows of SA | (N/WN) rows of SB ], each row
// sstr = round_up(K, 64) floats (the fused kernel streams 64-k strips with a
// 4-B/lane global_load_lds and must not cross rows).
inline int abft_sstr(int K) { return (K + 63) & ~63; }

template <int WM, int WN>
size_t abft_workspace_floats_t(int M, int N, int K) {
  // plain + row-weighted A-segment sums (B-side sums are not needed by the
  // ratio-locate scheme)
  return 2 * (size_t)(M / WM) * abft_sstr(K);
}

template <int BM, int BN, int BK, int WM, int WN, int MM>
hipError_t launch_tier(bool abft, bool inject, int M, int N, int K,
                       const float* A, const float* B, float* C, float alpha,
                       float beta, float tau, float inj_mag,
                       int verify_windows, float* ws, hipStream_t stream) {
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
  const float* SA = nullptr;
  int sstr = 0;
  if (abft) {
    // Precompute the plain + row-weighted segment checksums of A (per
    // WM-row band): one coalesced pass over the operand, ~1-2% of the GEMM
    // at N=4096 (vs -11.6% for the in-kernel cooperative sums pass this
    // replaces — tools/probe_ablate.hip).
    if (!ws) return hipErrorInvalidValue;
    sstr = abft_sstr(K);
    hipLaunchKernelGGL((segsum_kernel<WM>), dim3(K), dim3(256), 0, stream, M,
                       K, sstr, A, ws, ws + (size_t)(M / WM) * sstr);
    SA = ws;
  }
  if (abft && inject) {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BK, WM, WN, MM, true, true>), grid,
                       block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag, SA, sstr);
  } else if (abft) {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BK, WM, WN, MM, true, false>),
                       grid, block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag, SA, sstr);
  } else {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BK, WM, WN, MM, false, false>),
                       grid, block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag, SA, sstr);
  }
  return hipGetLastError();
}

}  // namespace ftsgemm
