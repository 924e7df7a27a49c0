// tier_launch.hpp — shared launcher template; each tier is compiled in its
// own translation unit (csrc/generated/kernel_<tier>.hip) so the register
// allocator sees one tier's three variants at a time (co-compiled template
// instantiations perturb each other's codegen — measured 88 vs 112 TF on
// the fused-ABFT huge tier in a monolithic TU).
#pragma once

#include <hip/hip_runtime.h>
#include <roctracer/roctx.h>

#include <cstdio>
#include <cstdlib>

#include "ft_kernels.hpp"
#include "ft_streamk.hpp"

namespace ftsgemm {

// roctx phase ranges (SURVEY.md §5 tracing row / VERDICT r01 next #9):
// every launch phase is bracketed so `rocprofv3 --marker-trace
// --kernel-trace` shows encode / GEMM / verify spans by name.  roctx is a
// no-op unless a tracer is attached.
struct RoctxRange {
  explicit RoctxRange(const char* name) { roctxRangePush(name); }
  ~RoctxRange() { roctxRangePop(); }
};

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
template <int BM, int BN, int BK, int BKF, int WM, int WN, int MM>
hipError_t launch_tier_streamk_t(bool abft, bool inject, int M, int N, int K,
                                 const float* A, const float* B, float* C,
                                 float alpha, float beta, float tau,
                                 float inj_mag, int verify_windows, float* ws,
                                 hipStream_t stream) {
  const int mode = streamk_env_mode();
  if (mode == 0) return hipErrorNotSupported;
  if (M % BM || N % BN || K % 64 || K % (abft ? BKF : BK) || M % 4 ||
      N % 4)
    return hipErrorNotSupported;
  constexpr int THREADS = 64 * (BM / WM) * (BN / WN);

  const void* kfn;
  if (abft && inject)
    kfn = (const void*)&sgemm_mfma_streamk<BM, BN, BKF, WM, WN, MM, true,
                                           true>;
  else if (abft)
    kfn = (const void*)&sgemm_mfma_streamk<BM, BN, BKF, WM, WN, MM, true,
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
    // measured crossover (profiles/sk_probe r2 steady-state + CLI-protocol
    // sweeps, two-kernel fixup), huge tier.  Two usage regimes matter:
    // pipelined launches (torch/bench) amortise the per-launch fixup +
    // workspace costs and SK wins from 1024 up (+31..+98%); the
    // reference's sync-per-rep CLI protocol pays them serially, where
    // plain SK only wins in the tail-waste regime (tiles >= G/2, 3072+:
    // +5..28%) and fused SK wins from 1536 up.  Shipped gate follows the
    // sync-protocol data (the conservative one; FT_SGEMM_STREAMK=1
    // forces SK for pipelined callers):
    //   engage iff waste >= 15%, outside the ~[G/3, G/2) dead band
    //   (2560: classic 40%-fill beats the combine), and for the plain
    //   kernel only in the tail regime (2*tiles >= G).
    // maxblk > 2 (fine-grained tiers, e.g. large at 10 blocks/CU):
    // dynamic dispatch self-balances the many short blocks and SK
    // measured 15-20% WORSE at every size -> classic.
    // dead band tiles in [G/4, G/2): 2560 classic wins outright and 2048
    // SK is run-to-run bimodal (34k..94k GFLOPS across sweeps) -> classic
    const bool dead_band = (4 * tiles >= G0) && (2 * tiles < G0);
    if (maxblk > 2 || waste < 0.15f || dead_band ||
        (!abft && 2 * tiles < G0)) {
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

  // The owner-fixup workspace is stream-ordered (concurrent GEMMs on
  // different streams must not share it); hipMallocAsync is not legal
  // inside graph capture, so captured launches take the classic path.
  hipStreamCaptureStatus cap = hipStreamCaptureStatusNone;
  (void)hipStreamIsCapturing(stream, &cap);
  if (cap != hipStreamCaptureStatusNone) return hipErrorNotSupported;

  const float* SA = nullptr;
  int sstr = 0;
  if (abft) {
    if (!ws) return hipErrorInvalidValue;
    sstr = abft_sstr(K);
    RoctxRange rr("abft_encode_segsum");
    hipLaunchKernelGGL((segsum_kernel<WM>), dim3(K), dim3(256), 0, stream, M,
                       K, sstr, A, ws);
    SA = ws;
  }
  RoctxRange rr_main(abft ? (inject ? "ft_sgemm_streamk_abft_inject"
                                    : "ft_sgemm_streamk_abft")
                          : "sgemm_streamk_plain");
  // inject+verify cadence: one pass per `istride` strip windows, targeting
  // ~verify_windows groups per tile (reference: 20 per GEMM)
  const int upt = K >> 6;
  int istride = upt / (verify_windows > 0 ? verify_windows : 20);
  if (istride < 1) istride = 1;

  // split-tile partial slots: 2 per workgroup (head + tail partials).
  // One-time: raise the default mempool's release threshold so the pool
  // RETAINS this allocation across stream syncs — with the default
  // threshold (0) every sync released the 100-200 MB back to the OS and
  // each launch re-mapped it, which cost more than the whole GEMM at
  // N<=2048 in sync-per-call usage (CLI sweep: 1024 huge 19.5k steady
  // -> 8.5k with per-rep remaps).
  static const bool pool_init = [] {
    int dev = 0;
    hipMemPool_t pool = nullptr;
    if (hipGetDevice(&dev) == hipSuccess &&
        hipDeviceGetDefaultMemPool(&pool, dev) == hipSuccess) {
      uint64_t thr = ~0ull;
      (void)hipMemPoolSetAttribute(pool, hipMemPoolAttrReleaseThreshold,
                                   &thr);
    }
    return true;
  }();
  (void)pool_init;
  float* partials = nullptr;
  const size_t pbytes = (size_t)2 * G * BM * BN * sizeof(float);
  if (hipMallocAsync((void**)&partials, pbytes, stream) != hipSuccess)
    return hipErrorNotSupported;  // nothing mutated yet; classic path

  if (abft && inject) {
    hipLaunchKernelGGL(
        (sgemm_mfma_streamk<BM, BN, BKF, WM, WN, MM, true, true>), dim3(G),
        dim3(THREADS), 0, stream, M, N, K, A, B, C, alpha, beta, istride,
        tau, inj_mag, SA, sstr, partials);
  } else if (abft) {
    hipLaunchKernelGGL(
        (sgemm_mfma_streamk<BM, BN, BKF, WM, WN, MM, true, false>), dim3(G),
        dim3(THREADS), 0, stream, M, N, K, A, B, C, alpha, beta, istride,
        tau, inj_mag, SA, sstr, partials);
  } else {
    hipLaunchKernelGGL(
        (sgemm_mfma_streamk<BM, BN, BK, WM, WN, MM, false, false>), dim3(G),
        dim3(THREADS), 0, stream, M, N, K, A, B, C, alpha, beta, istride,
        tau, inj_mag, SA, sstr, partials);
  }
  hipError_t err = hipGetLastError();
  if (err == hipSuccess) {
    // combine split tiles (fully-owned tiles exit immediately)
    hipLaunchKernelGGL((sk_fixup_kernel<BM, BN, WM, WN, MM>), dim3(tiles),
                       dim3(THREADS), 0, stream, M, N, upt, G, alpha, beta,
                       C, partials);
    err = hipGetLastError();
  }
  (void)hipFreeAsync(partials, stream);
  return err;
}

// ABFT workspace layout: (M/WM) rows, one per WM-band of A, each of
// 2 * sstr floats holding the (plain, row-weighted) segment-sum pairs
// INTERLEAVED at [2k], [2k+1]; sstr = round_up(K, 64) so the fused
// kernel's 64-k strip loads never cross rows.  B-side sums are not needed
// by the ratio-locate scheme.
inline int abft_sstr(int K) { return (K + 63) & ~63; }

template <int WM, int WN>
size_t abft_workspace_floats_t(int M, int N, int K) {
  // plain + row-weighted A-segment sums (B-side sums are not needed by the
  // ratio-locate scheme)
  return 2 * (size_t)(M / WM) * abft_sstr(K);
}

// BKF: K-panel depth of the fused-ABFT twin (may be smaller than the
// plain BK so the +1KB/wave checksum strips don't push the block over an
// LDS-occupancy boundary — kernel_table.py bkf note).
template <int BM, int BN, int BK, int BKF, int WM, int WN, int MM>
hipError_t launch_tier(bool abft, bool inject, int M, int N, int K,
                       const float* A, const float* B, float* C, float alpha,
                       float beta, float tau, float inj_mag,
                       int verify_windows, float* ws, hipStream_t stream) {
  const int bk_used = abft ? BKF : BK;
  if (M % BM || N % BN || K % bk_used || M % 4 || N % 4)
    return hipErrorInvalidValue;
  dim3 grid(M / BM, N / BN);
  dim3 block(64 * (BM / WM) * (BN / WN));
  const int niter = K / bk_used;
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
    RoctxRange rr("abft_encode_segsum");
    hipLaunchKernelGGL((segsum_kernel<WM>), dim3(K), dim3(256), 0, stream, M,
                       K, sstr, A, ws);
    SA = ws;
  }
  RoctxRange rr_main(abft ? (inject ? "ft_sgemm_abft_inject" : "ft_sgemm_abft")
                          : "sgemm_plain");
  if (abft && inject) {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BKF, WM, WN, MM, true, true>), grid,
                       block, 0, stream, M, N, K, A, B, C, alpha, beta,
                       stride, stride, tau, inj_mag, SA, sstr);
  } else if (abft) {
    hipLaunchKernelGGL((sgemm_mfma<BM, BN, BKF, WM, WN, MM, true, false>),
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
