// ft_kernels.hpp — MI355X (gfx950/CDNA4) fault-tolerant SGEMM kernel family.
//
// Brand-new design with the capabilities of the reference CUDA kernel library
// (/root/reference/kernel/ft_sgemm/include_code_gen/*.cuh, algorithm traced
// in SURVEY.md §2.3), re-architected for CDNA4:
//
//   * f32-input MFMA outer products (v_mfma_f32_32x32x2_f32 for the
//     32x32-fragment tiers, v_mfma_f32_16x16x4_f32 for the small tier) —
//     exact fp32 numerics at the 157 TF/s f32 vector rate.
//   * LDS double-buffered A/B K-panels staged with global_load_lds
//     (async HBM->LDS DMA, 16 B per lane), one __syncthreads per K-panel.
//   * Checksum ENCODE from precomputed segment sums: a separate
//     bandwidth-bound kernel (segsum_kernel) computes, per WM-row band of
//     A, the plain and the row-index-weighted column sums in one coalesced
//     pass (~1-2% of GEMM time at N=4096); the fused kernel streams each
//     wave's 64-k strip of them into a private LDS strip with one
//     4-B-per-lane global_load_lds per panel pair.  This replaces the
//     reference's per-k-step in-loop shuffle encode
//     (ft_sgemm_huge.cuh:150-213) and the earlier in-kernel cooperative
//     panel-sums pass: ablation (tools/probe_ablate.hip) measured the
//     in-kernel sums pass + its extra barrier at -11.6% of plain GEMM
//     throughput, vs ~2% for the precompute pass.
//   * 64-lane-wavefront ABFT verify with RATIO LOCATE: each wave maintains
//     per-lane column checksums (plain cc + row-weighted cw) of its output
//     sub-tile.  A fault's column is lane-local (MFMA acc column == lane %
//     MM), its magnitude is the plain column residual rc, and its row is
//     round(rw/rc) — no cross-lane row reductions at all, unlike the
//     reference's block-wide LDS transpose-reduce + row x col intersection
//     (ft_sgemm_huge.cuh:346-485).
//   * periodic in-kernel verify -> ratio locate -> branch-free in-register
//     correction, and a deterministic rotating fault injector (template
//     flag, not hard-coded: SURVEY.md §5 asks for measurable overhead with
//     and without injection).
//
// Matrix semantics (reference parity, sgemm.cu:108): C = alpha*A*B^T + beta*C,
// A MxK / B NxK / C MxN, all column-major.  Host launchers require
// M % BM == 0, N % BN == 0, K % BK == 0 and M,N multiples of 4 (the sweep
// sizes 1024..6144 step 512 and the distributed N=32768 all satisfy this).

#pragma once

#include <hip/hip_runtime.h>

// FT_PARANOID (race-check build, `make cli-paranoid`): drain ALL
// outstanding async staging (glds vmcnt + LDS lgkm) and barrier right at
// every staging site.  Per-thread arithmetic order is unchanged, so the
// output must be BIT-IDENTICAL to the normal build; any difference is a
// staging race (SURVEY.md §5 race-detection row).
#ifdef FT_PARANOID
#define FT_PARA_SYNC()                                                  \
  do {                                                                  \
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");         \
    __syncthreads();                                                    \
  } while (0)
#else
#define FT_PARA_SYNC() \
  do {                 \
  } while (0)
#endif

namespace ftsgemm {

using f32x2 = __attribute__((ext_vector_type(2))) float;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;

template <int MM> struct mfma_traits;

// v_mfma_f32_32x32x2_f32: lane l holds A[i=l&31][k=l>>5], B[k=l>>5][j=l&31];
// C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), reg in [0,16).
template <> struct mfma_traits<32> {
  static constexpr int kstep = 2;
  static constexpr int nreg = 16;
  using acc_t = f32x16;
  static __device__ inline acc_t mma(float a, float b, acc_t c) {
    return __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, c, 0, 0, 0);
  }
};

// v_mfma_f32_16x16x4_f32: lane l holds A[i=l&15][k=l>>4], B[k=l>>4][j=l&15];
// C/D: col = lane&15, row = (lane>>4)*4 + reg, reg in [0,4).
template <> struct mfma_traits<16> {
  static constexpr int kstep = 4;
  static constexpr int nreg = 4;
  using acc_t = f32x4;
  static __device__ inline acc_t mma(float a, float b, acc_t c) {
    return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
  }
};

// Completes a per-k-slice partial into a full sum across the 64/MM slices
// (masks MM..32).
template <int MM> __device__ inline float slice_sum(float v) {
#pragma unroll
  for (int m = MM; m < 64; m <<= 1) v += __shfl_xor(v, m, 64);
  return v;
}

// Double-precision variant for the cold locate path (shuffles the two
// 32-bit halves).
template <int MM> __device__ inline double dslice_sum(double v) {
#pragma unroll
  for (int m = MM; m < 64; m <<= 1) {
    union { double d; int i[2]; } u{v};
    u.i[0] = __shfl_xor(u.i[0], m, 64);
    u.i[1] = __shfl_xor(u.i[1], m, 64);
    v += u.d;
  }
  return v;
}

// Accumulator register -> row within the MM x MM fragment (valid for both
// MFMA shapes used here; for MM=16, reg>>2 == 0).
__device__ constexpr int acc_row(int reg, int sub) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * sub;
}

// Segment-sum precompute: for a column-major MxK matrix, per SEG-row band,
// the plain column sum s = sum_{i in [seg*SEG, +SEG)} A[i + k*M] and the
// row-index-weighted sum w = sum_i (i % SEG) * A[i + k*M] used by the
// ratio locate (fault row = round(rw / rc)).  The pair is written
// INTERLEAVED — S[seg * 2*sstr + 2k] = s, S[seg * 2*sstr + 2k + 1] = w —
// so the fused kernel's encode fetches both with a single ds_read_b64 per
// k-step (the two separate b32 broadcast reads were one of the extra
// issue slots behind the tall tier's 21% fused overhead).
// One workgroup per column k, coalesced f32x4 sweep, SEG/4-lane shuffle
// groups, no LDS, no atomics.  sstr must be >= K and a multiple of 64 (the
// fused kernel streams 64-k strips with 4-B/lane global_load_lds pairs,
// never crossing rows).
template <int SEG>
__global__ __launch_bounds__(256) void segsum_kernel(
    int M, int K, int sstr, const float* __restrict__ A,
    float* __restrict__ S) {
  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  const float* col = A + (size_t)k * M;
  constexpr int GL = SEG / 4;  // lanes per segment group (4|8|16)
  for (int base = 0; base < M; base += 1024) {
    const int idx = base + tid * 4;
    float s = 0.f, w = 0.f;
    if (idx < M) {
      const f32x4 v = *(const f32x4*)(col + idx);
      s = (v[0] + v[1]) + (v[2] + v[3]);
      const float i0 = (float)(idx % SEG);
      w = i0 * v[0] + (i0 + 1.f) * v[1] + (i0 + 2.f) * v[2] +
          (i0 + 3.f) * v[3];
    }
#pragma unroll
    for (int m = 1; m < GL; m <<= 1) {
      s += __shfl_xor(s, m, 64);
      w += __shfl_xor(w, m, 64);
    }
    if (idx < M && (idx % SEG) == 0) {
      S[(size_t)(idx / SEG) * 2 * sstr + 2 * k] = s;
      S[(size_t)(idx / SEG) * 2 * sstr + 2 * k + 1] = w;
    }
  }
}

// NTC: non-temporal C epilogue loads/stores (beta != 0 streams 2x64 MB of C
// through the caches per GEMM at N=4096 — nt keeps it from evicting the
// L3-resident A/B panels).  SWIZ: bijective XCD-aware blockIdx remap.
// PIPE=1 (plain kernels only): 3-LDS-buffer glds ring with raw s_barrier +
// counted s_waitcnt — panels it+1 and it+2 stay in flight across the
// barrier instead of draining at every __syncthreads.
template <int BM, int BN, int BK, int WM, int WN, int MM, bool ABFT,
          bool INJECT, bool NTC = false, bool SWIZ = false, int OCC = 2,
          int PIPE = 0>
__global__ __launch_bounds__(64 * (BM / WM) * (BN / WN), OCC) void sgemm_mfma(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta, int verify_iters, int inject_stride, float tau, float inj_mag,
    const float* __restrict__ SA, int sstr) {
  using T = mfma_traits<MM>;
  constexpr int KSTEP = T::kstep;
  constexpr int NREG = T::nreg;
  constexpr int WAVES_M = BM / WM, WAVES_N = BN / WN;
  constexpr int NWAVES = WAVES_M * WAVES_N;
  constexpr int THREADS = NWAVES * 64;
  constexpr int FM = WM / MM, FN = WN / MM;
  constexpr int BUF = (BM + BN) * BK;  // floats per double-buffer half
  // ABFT checksum strips: per wave, a private [2 pair-buffers][sa(64) |
  // saw(64)] window of the precomputed segment sums, streamed by
  // global_load_lds.  Only A-side sums are needed: the column checksums
  // detect, the weighted ones locate the row, and the column is lane-local.
  static_assert(!(ABFT && PIPE) && !(INJECT && PIPE),
                "3-buffer ring is plain-only");
  constexpr int NBUF = PIPE ? 3 : 2;
  constexpr int STRIP_OFF = NBUF * BUF;
  constexpr int LDS_FLOATS = ABFT ? (STRIP_OFF + NWAVES * 256)
                                  : (NBUF * BUF);

  __shared__ __attribute__((aligned(16))) float lds[LDS_FLOATS];

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane / MM;  // k-slice within one MFMA k-step
  const int r = lane % MM;    // row index of A fragment / col index of B
  const int wm_idx = wave / WAVES_N, wn_idx = wave % WAVES_N;
  const int wi0 = wm_idx * WM;
  const int wj0 = wn_idx * WN;
  int bx = blockIdx.x, by = blockIdx.y;
  if constexpr (SWIZ) {
    // bijective XCD remap: consecutive remapped ids land on one XCD so
    // each XCD's L2 sees a contiguous band of output tiles.
    const int nwg = gridDim.x * gridDim.y;
    const int orig = by * gridDim.x + bx;
    const int qq = nwg / 8, rr = nwg % 8;
    const int xcd = orig % 8;
    const int wgid =
        (xcd < rr ? xcd * (qq + 1) : rr * (qq + 1) + (xcd - rr) * qq) +
        orig / 8;
    bx = wgid % gridDim.x;
    by = wgid / gridDim.x;
  }
  const int im0 = bx * BM;
  const int jn0 = by * BN;

  typename T::acc_t acc[FM][FN] = {};
  float cc[FN] = {};  // running column checksum of this wave's tile (per
                      // lane: col r of frag fn, this lane's k-slices only)
  float cw[FN] = {};  // row-index-weighted column checksum (ratio locate:
                      // fault row = round(residual(cw) / residual(cc)))

  // ---- async HBM -> LDS staging (global_load_lds, 16 B per lane) ----
  // staging passes may be fractional in WHOLE WAVES (e.g. the fused tall
  // twin at BKF=8 stages a 256-float B panel with a 128-thread block:
  // GB=0 full passes + a 64-lane remainder by wave 0)
  constexpr int GA = (BM * BK) / (THREADS * 4);  // dwordx4 chunks for A
  constexpr int GAR = (BM * BK / 4) % THREADS;   // remainder lanes
  constexpr int GB = (BN * BK) / (THREADS * 4);
  constexpr int GBR = (BN * BK / 4) % THREADS;
  static_assert(GAR % 64 == 0 && GBR % 64 == 0,
                "stage remainder must be whole waves");
  static_assert(GA + GAR > 0 && GB + GBR > 0, "tile too small");

  auto stage = [&](int q, int k0) __attribute__((always_inline)) {
    float* dstA = &lds[q * BUF];
    float* dstB = &lds[q * BUF + BM * BK];
#pragma unroll
    for (int t = 0; t < GA; ++t) {
      const int f = (t * THREADS + tid) * 4;  // per-lane float index
      const int k = f / BM, i = f % BM;       // LDS layout [k][i], i contig
      const float* g = A + (im0 + i) + (size_t)(k0 + k) * M;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)g,
          (__attribute__((address_space(3))) void*)(dstA +
                                                    (t * THREADS + wave * 64) *
                                                        4),
          16, 0, 0);
    }
    if constexpr (GAR > 0) {
      if (tid < GAR) {
        const int f = (GA * THREADS + tid) * 4;
        const int k = f / BM, i = f % BM;
        const float* g = A + (im0 + i) + (size_t)(k0 + k) * M;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)g,
            (__attribute__((address_space(3))) void*)(dstA +
                                                      (GA * THREADS +
                                                       wave * 64) *
                                                          4),
            16, 0, 0);
      }
    }
#pragma unroll
    for (int t = 0; t < GB; ++t) {
      const int f = (t * THREADS + tid) * 4;
      const int k = f / BN, j = f % BN;
      const float* g = B + (jn0 + j) + (size_t)(k0 + k) * N;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)g,
          (__attribute__((address_space(3))) void*)(dstB +
                                                    (t * THREADS + wave * 64) *
                                                        4),
          16, 0, 0);
    }
    if constexpr (GBR > 0) {
      if (tid < GBR) {
        const int f = (GB * THREADS + tid) * 4;
        const int k = f / BN, j = f % BN;
        const float* g = B + (jn0 + j) + (size_t)(k0 + k) * N;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)g,
            (__attribute__((address_space(3))) void*)(dstB +
                                                      (GB * THREADS +
                                                       wave * 64) *
                                                          4),
            16, 0, 0);
      }
    }
  };

  // ---- ABFT strip staging: one 4-B/lane glds pair per TWO K panels ----
  // Each wave streams its own 64-k window of the precomputed segment sums
  // into a private LDS strip.  No extra barrier, no cooperative pass: the
  // strips ride the same prefetch pipeline as the A/B panels and are
  // drained by the same end-of-panel __syncthreads.
  // Workspace layout (set up by the launcher): per WM-row band of A, one
  // row of 2*sstr floats with (plain, row-weighted) segment sums
  // INTERLEAVED at [2k], [2k+1] — a 64-k window is 128 contiguous floats,
  // fetched by two 64-lane glds loads and consumed with one ds_read_b64
  // per k-step.
  const int segA = bx * WAVES_M + wm_idx;
  auto strip_stage = [&](int pb, int k0) __attribute__((always_inline)) {
    // glds source addresses are PER-LANE (the LDS side is uniform base +
    // lane*size): lane l fetches pair-element (2*k0 + l).
    const float* ga = SA + (size_t)segA * 2 * sstr + 2 * k0 + lane;
    float* dst = &lds[STRIP_OFF + wave * 256 + pb * 128];
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)ga,
        (__attribute__((address_space(3))) void*)dst, 4, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(ga + 64),
        (__attribute__((address_space(3))) void*)(dst + 64), 4, 0, 0);
  };

  // ---- ABFT verify / locate / correct: wave-autonomous, registers only ----
  // Two-phase: a cheap detect (compare the total tile sum against the total
  // checksum — FM*FN*NREG adds + a butterfly) runs every verify window; the locate/correct
  // below is entered only when the residual trips the threshold (i.e. in
  // the wave that actually absorbed a fault), so the fault-free common case
  // never pays for location.
  //
  // Ratio locate: the fault's COLUMN is lane-local (acc column == lane's
  // r), its magnitude is the column residual rc; its ROW index is
  // round(rw / rc) where rw is the residual of the row-index-WEIGHTED
  // column checksum.  This replaces the reference's row x column residual
  // intersection (ft_sgemm_huge.cuh:422-485) whose row residuals needed a
  // full cross-lane butterfly per accumulator register — the measured
  // window cost dropped ~10x because no row reductions exist at all.
  //
  // Guarantee (same as the reference): ONE fault per verify window per
  // wave tile.  TWO faults e1 at row r1 and e2 at row r2 landing in the
  // SAME column of one wave tile within one window degrade differently
  // from the reference: rc = e1 + e2, row = round((r1*e1 + r2*e2) /
  // (e1+e2)), and rc is subtracted IF some accumulator row matches that
  // blended index — correcting a clean element by the combined magnitude
  // (the reference's intersection scheme instead corrects every
  // row-candidate x col-candidate).  When the blend rounds outside the
  // fragment's row range nothing matches and the kernel fail-safes to
  // detected-but-uncorrected.  Faults in different columns (the common
  // double-fault shape) locate independently per lane and correct fine.
  auto locate_correct = [&]() __attribute__((always_inline)) {
    // Opaque copy of `sub`: everything derived from it (the 2*FM*NREG
    // per-lane row weights below) is materialised INSIDE this cold block.
    // Without it the compiler hoists the loop-invariant weights into the
    // kernel prologue and keeps them live through the whole hot loop
    // (+68 VGPRs measured on the 64x64-wave tile -> spills on the bigger
    // tiles).
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      // fp64 PLAIN column sum: the fault itself (|e| ~ 1e4) sits in this
      // sum, so an fp32 accumulation rounds every later term at ulp(1e4)
      // ~ 1e-3 and the CORRECTION (rc is subtracted from the output)
      // inherits ~5e-3..1e-2 of error (found by tools/soak.py at
      // |alpha| > 1).  The WEIGHTED sum stays fp32: rw only selects an
      // integer row via round(rw/rc), its baseline cw is fp32-maintained
      // anyway, and the fp32 noise is ~1e-4 of a row index — while the
      // fp64 weight conversions were what spilled ~143 dwords/lane in the
      // stream-K fused variants (cold-path scratch traffic ~1 GB/GEMM).
      int sub_o = sub;  // fresh opaque copy per fn: stops the
      asm volatile("" : "+v"(sub_o));  // weights being CSE'd+spilled
      double colp = 0.0;
      float colwf = 0.f;
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int reg = 0; reg < NREG; ++reg) {
          const float v = acc[fm][fn][reg];
          colp += (double)v;
          colwf = fmaf((float)(fm * MM + acc_row(reg, sub_o)), v, colwf);
        }
      const float rc =
          (float)(dslice_sum<MM>(colp) - (double)slice_sum<MM>(cc[fn]));
      const float rw = slice_sum<MM>(colwf) - slice_sum<MM>(cw[fn]);
      const bool cbad = fabsf(rc) > tau;
      const int row = (int)rintf(rw / (cbad ? rc : 1.f));
      int sub_c = sub;  // independent opaque copy for the correct pass
      asm volatile("" : "+v"(sub_c));
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int reg = 0; reg < NREG; ++reg) {
          const bool hit = cbad && (fm * MM + acc_row(reg, sub_c) == row);
          acc[fm][fn][reg] -= hit ? rc : 0.f;
        }
    }
  };

  auto verify_correct = [&]() __attribute__((always_inline)) {
    // Cheap detect: total tile sum vs total checksum (FM*FN*NREG adds + 6
    // shuffles).  The residual is identical across the wave's lanes, so
    // the branch is uniform; only a wave that actually absorbed a fault
    // enters the locate/correct path above.
    float tot = 0.f, chk = 0.f;
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      chk += cc[fn];
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int reg = 0; reg < NREG; ++reg) tot += acc[fm][fn][reg];
    }
    float res = tot - chk;
#pragma unroll
    for (int m = 1; m < 64; m <<= 1) res += __shfl_xor(res, m, 64);
    if (__builtin_expect(fabsf(res) > tau, 0)) locate_correct();
  };

  // ---- main K loop: one barrier per BK panel, glds prefetch overlaps ----
  // Structure: bursts of `verify_iters` panels.  The inner panel loop never
  // touches the accumulator with VALU code; injection and verify/correct
  // run at burst boundaries only (a conditional VALU read of acc inside the
  // panel loop makes hipcc shuttle the whole accumulator file through VGPR
  // copies every panel — measured 2x wall time on the huge tier).
  // A strip load covers 64 k values = PPS panels; strips are double
  // buffered by strip-window parity.
  constexpr int PPS = 64 / BK;  // panels per strip window (BK <= 64)
  static_assert(64 % BK == 0 || !ABFT, "ABFT needs BK dividing 64");
  constexpr int GLD = GA + GB;  // glds instructions per panel
  stage(0, 0);
  if constexpr (ABFT) strip_stage(0, 0);  // window (panels 0..PPS-1)
  FT_PARA_SYNC();
  if constexpr (PIPE) {
    // 3-buffer ring: prologue stages two panels; the main loop keeps the
    // newest one in flight across each barrier.
    if (BK < K) stage(1, BK);
  }
  if constexpr (!PIPE) __syncthreads();  // drains in-flight glds

  const int niter = K / BK;
  if constexpr (PIPE) {
    for (int it = 0; it < niter; ++it) {
      const int q = it % 3;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();  // readers of buf (it+2)%3 are done
      if (it + 2 < niter) {
        stage((it + 2) % 3, (it + 2) * BK);
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * GLD) : "memory");
      } else if (it + 1 < niter) {
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(GLD) : "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      const float* As = &lds[q * BUF];
      const float* Bs = &lds[q * BUF + BM * BK];
#pragma unroll
      for (int kk = 0; kk < BK / KSTEP; ++kk) {
        const int kloc = kk * KSTEP + sub;
        float a[FM], b[FN];
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
          a[fm] = As[kloc * BM + wi0 + fm * MM + r];
#pragma unroll
        for (int fn = 0; fn < FN; ++fn)
          b[fn] = Bs[kloc * BN + wj0 + fn * MM + r];
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
#pragma unroll
          for (int fn = 0; fn < FN; ++fn)
            acc[fm][fn] = T::mma(a[fm], b[fn], acc[fm][fn]);
      }
    }
  }
  int it = PIPE ? niter : 0;  // PIPE path already done; skip burst loop
  while (it < niter) {
    if constexpr (INJECT) {
      // Deterministic rotating injector, once per verify window (reference:
      // ft_sgemm_huge.cuh:324-327 injects in every block with a rotating
      // thread id; here the victim also rotates across waves).
      if (tid == ((unsigned)(it / inject_stride) * 67u) % THREADS)
        acc[0][0][0] += inj_mag;
    }
    const int burst_end = (it + verify_iters < niter) ? it + verify_iters
                                                      : niter;
    for (; it < burst_end; ++it) {
      const int q = it & 1;
      if (it + 1 < niter) {
        stage(q ^ 1, (it + 1) * BK);
        // a new strip window rides along every PPS-th panel prefetch
        if constexpr (ABFT) {
          if ((it + 1) % PPS == 0) strip_stage(((it + 1) / PPS) & 1,
                                               (it + 1) * BK);
        }
      }
      FT_PARA_SYNC();
      const float* As = &lds[q * BUF];
      const float* Bs = &lds[q * BUF + BM * BK];
      const float* strip =
          ABFT ? &lds[STRIP_OFF + wave * 256 + ((it / PPS) & 1) * 128 +
                      (it % PPS) * BK * 2]
               : nullptr;
#pragma unroll
      for (int kk = 0; kk < BK / KSTEP; ++kk) {
        const int kloc = kk * KSTEP + sub;
        float a[FM], b[FN];
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
          a[fm] = As[kloc * BM + wi0 + fm * MM + r];
#pragma unroll
        for (int fn = 0; fn < FN; ++fn)
          b[fn] = Bs[kloc * BN + wj0 + fn * MM + r];

        if constexpr (ABFT) {
          // Encode: ONE ds_read_b64 fetches the interleaved (plain,
          // row-weighted) segment-sum pair for this k-slice + two fmas
          // per B fragment into the running column checksums (reference
          // encode: ft_sgemm_huge.cuh:150-213, redesigned around the
          // offline segsum pass + ratio locate).
          const f32x2 sw = *(const f32x2*)(strip + 2 * kloc);
#pragma unroll
          for (int fn = 0; fn < FN; ++fn) {
            cc[fn] = fmaf(sw[0], b[fn], cc[fn]);
            cw[fn] = fmaf(sw[1], b[fn], cw[fn]);
          }
        }

        // interleave VMEM/DS/MFMA scheduling groups (+1% measured)
        __builtin_amdgcn_iglp_opt(0);
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
#pragma unroll
          for (int fn = 0; fn < FN; ++fn)
            acc[fm][fn] = T::mma(a[fm], b[fn], acc[fm][fn]);
      }
      __syncthreads();
    }
    if constexpr (ABFT) verify_correct();
    if constexpr (!ABFT && !INJECT) { /* plain: single burst, no verify */ }
  }

  // ---- epilogue: alpha/beta merge, float4 along column-major columns ----
#pragma unroll
  for (int fm = 0; fm < FM; ++fm)
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      const int j = jn0 + wj0 + fn * MM + r;
      float* colbase = C + (size_t)j * M + im0 + wi0 + fm * MM;
#pragma unroll
      for (int g = 0; g < NREG / 4; ++g) {
        float* p = colbase + 4 * sub + 8 * g;
        f32x4 out;
        if (beta != 0.f) {
          const f32x4 prev = NTC ? __builtin_nontemporal_load((const f32x4*)p)
                                 : *(const f32x4*)p;
#pragma unroll
          for (int u = 0; u < 4; ++u)
            out[u] = alpha * acc[fm][fn][4 * g + u] + beta * prev[u];
        } else {
#pragma unroll
          for (int u = 0; u < 4; ++u) out[u] = alpha * acc[fm][fn][4 * g + u];
        }
        if constexpr (NTC)
          __builtin_nontemporal_store(out, (f32x4*)p);
        else
          *(f32x4*)p = out;
      }
    }
}

}  // namespace ftsgemm
