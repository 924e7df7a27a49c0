// tier_launch.hpp — shared launcher template; each tier is compiled in its
// own translation unit (csrc/generated/kernel_<tier>.hip) so the register
// allocator sees one tier's three variants at a time (co-compiled template
// instantiations perturb each other's codegen — measured 88 vs 112 TF on
// the fused-ABFT huge tier in a monolithic TU).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#include "ft_kernels.hpp"
#include "ft_streamk.hpp"

namespace ftsgemm {

// FT_SGEMM_STREAMK: 0 = never, 1 = force (when the shape divides),
// unset/other = auto (use stream-K when the classic tile-per-workgroup
// grid would waste >15% of a dispatch round on tail quantization).
inline int streamk_env_mode() {
  // read fresh each launch (cheap vs a kernel launch) so tests can toggle
  // the mode inside one process
  const char* e = std::getenv("FT_SGEMM_STREAMK");
  return e ? std::atoi(e) : 2;
}

inline int abft_sstr(int K);  // defined below

inline int device_cu_count() {
  static int cus = [] {
    int dev = 0, n = 0;
    (void)hipGetDevice(&dev);
    (void)hipDeviceGetAttribute(&n, hipDeviceAttributeMultiprocessorCount,
                                dev);
    return n > 0 ? n : 256;
  }();
  return cus;
}

// Stream-K launch of one tier.  Returns hipErrorNotSupported when the
// classic launch is preferable (heuristic) or the shape does not divide —
// the caller then falls through to launch_tier.
template <int BM, int BN, int BK, int WM, int WN, int MM>
hipError_t launch_tier_streamk_t(bool abft, bool inject, int M, int N, int K,
                                 const float* A, const float* B, float* C,
                                 float alpha, float beta, float tau,
                                 float inj_mag, int verify_windows, float* ws,
                                 hipStream_t stream) {
  const int mode = streamk_env_mode();
  if (mode == 0) return hipErrorNotSupported;
  if (M % BM || N % BN || K % 64 || K % BK || M % 4 || N % 4)
    return hipErrorNotSupported;
  constexpr int THREADS = 64 * (BM / WM) * (BN / WN);

  const void* kfn;
  if (abft && inject)
    kfn = (const void*)&sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, true,
                                           true>;
  else if (abft)
    kfn = (const void*)&sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, true,
                                           false>;
  else
    kfn = (const void*)&sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, false,
                                           false>;
  int maxblk = 0;
  const hipError_t oe =
      hipOccupancyMaxActiveBlocksPerMultiprocessor(&maxblk, kfn, THREADS, 0);
  const bool dbg = std::getenv("FT_SGEMM_SK_DEBUG") != nullptr;
  if (oe != hipSuccess || maxblk < 1) {
    if (dbg)
      fprintf(stderr, "[sk %dx%d] occupancy query failed: err=%d maxblk=%d\n",
              BM, BN, (int)oe, maxblk);
    return hipErrorNotSupported;
  }

  const int tiles = (M / BM) * (N / BN);
  int G0 = device_cu_count() * maxblk;
  const char* ge = std::getenv("FT_SGEMM_SK_G");  // tuning override
  if (ge) G0 = std::atoi(ge);
  if (G0 < 1) return hipErrorNotSupported;
  if (mode != 1) {
    const float rounds = ceilf((float)tiles / (float)G0);
    const float waste = (rounds * G0 - tiles) / (rounds * G0);
    if (waste < 0.15f) {
      if (dbg)
        fprintf(stderr, "[sk %dx%d] M=%d N=%d: classic (waste %.3f)\n", BM,
                BN, M, N, waste);
      return hipErrorNotSupported;
    }
  }
  const int total = tiles * (K >> 6);
  const int G = total < G0 ? total : G0;
  if (dbg)
    fprintf(stderr,
            "[sk %dx%d] M=%d N=%d K=%d: ENGAGED tiles=%d maxblk=%d G=%d "
            "units/wg~%.1f\n",
            BM, BN, M, N, K, tiles, maxblk, G, (float)total / G);

  // beta applied once up front (split tiles accumulate into C)
  const size_t total4 = (size_t)M * N / 4;
  const int pgrid =
      (int)((total4 + 255) / 256 < 8192 ? (total4 + 255) / 256 : 8192);
  hipLaunchKernelGGL(prescale_kernel, dim3(pgrid), dim3(256), 0, stream,
                     total4, beta, C);

  const float* SA = nullptr;
  int sstr = 0;
  if (abft) {
    if (!ws) return hipErrorInvalidValue;
    sstr = abft_sstr(K);
    hipLaunchKernelGGL((segsum_kernel<WM>), dim3(K), dim3(256), 0, stream, M,
                       K, sstr, A, ws, ws + (size_t)(M / WM) * sstr);
    SA = ws;
  }
  // inject+verify cadence: one pass per `istride` strip windows, targeting
  // ~verify_windows groups per tile (reference: 20 per GEMM)
  const int upt = K >> 6;
  int istride = upt / (verify_windows > 0 ? verify_windows : 20);
  if (istride < 1) istride = 1;

  if (abft && inject) {
    hipLaunchKernelGGL(
        (sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, true, true>), dim3(G),
        dim3(THREADS), 0, stream, M, N, K, A, B, C, alpha, istride, tau,
        inj_mag, SA, sstr);
  } else if (abft) {
    hipLaunchKernelGGL(
        (sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, true, false>), dim3(G),
        dim3(THREADS), 0, stream, M, N, K, A, B, C, alpha, istride, tau,
        inj_mag, SA, sstr);
  } else {
    hipLaunchKernelGGL(
        (sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, false, false>), dim3(G),
        dim3(THREADS), 0, stream, M, N, K, A, B, C, alpha, istride, tau,
        inj_mag, SA, sstr);
  }
  return hipGetLastError();
}

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
