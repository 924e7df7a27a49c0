// ft_streamk.hpp — stream-K work decomposition for the CDNA4 SGEMM family
// (plain + fused-ABFT), fixing the grid-straggler sizes of the classic
// tile-per-workgroup launch (VERDICT r01 missing #2 / next #3).
//
// Problem: the classic launch runs (M/BM)*(N/BN) workgroups; MI355X holds
// 512 co-resident huge-tier blocks (256 CUs x 2), so e.g. N=3072 (288
// tiles) runs one full round plus a 32-block tail round — measured 74.6 TF
// vs 132.5 at N=4096.  The reference never faces this (its T4 has 40 SMs;
// every sweep size over-fills it), but its ratio-to-vendor-BLAS is >=100%
// at EVERY size (README.md:38-53), so ours must not crater.
//
// Design (work-centric, CUTLASS-stream-K-like but built around the ABFT
// strip windows): the GEMM is a list of `total = ntiles * (K/64)` work
// units — one unit = one 64-k strip window of one output tile (4 BK=16
// LDS panels, exactly the granularity the ABFT checksum strips are staged
// at).  A fixed grid of G workgroups (G = CUs x occupancy, never more
// than total — ALL workgroups are co-resident, which the fixup protocol
// relies on) each processes a CONTIGUOUS unit range, k-fastest: perfect
// load balance at any size, no tail round.
//
// Split-tile combine is a deterministic TWO-KERNEL reduce:
//   * a tile fully inside one workgroup's range takes the normal
//     alpha/beta epilogue directly in the main kernel;
//   * split-tile segments write their raw partial accumulator to private
//     workspace slots (head partial -> slot 2g, tail partial -> 2g+1) —
//     coalesced streaming stores, no contention;
//   * sk_fixup_kernel (launched after, one workgroup per tile, identical
//     thread geometry) sums the slots of a split tile in fixed workgroup
//     order and applies the alpha/beta epilogue; fully-owned tiles exit
//     in a few instructions.
// Two rejected designs, both measured: an atomic f32 combine was 3x
// SLOWER than the classic straggler grid at N=1024 (16 workgroups
// hammering the same 128 KB of C serialize in L2), and an in-kernel
// owner-spins-on-contributor-flags protocol deadlocked whenever the
// dispatcher did not make the whole grid co-resident (fresh-process CLI
// runs dispatched partially and hung at N=4608; the same binary was fine
// after prior in-process GPU activity).  The two-kernel form needs no
// fences, no atomics and no residency assumption: kernel-boundary
// ordering publishes the partials.
//
// ABFT composes cleanly: each workgroup's per-lane column checksums (cc,
// cw) cover exactly the k-range it accumulated, so the wave-autonomous
// verify -> ratio-locate -> correct runs unchanged on the partial
// products, once per `istride` strip windows and at segment end (every
// injection is verified+corrected before the tile contribution leaves the
// registers).  The injector keeps the reference's rotating-victim
// self-test semantics per (tile, window-group).
#pragma once

#include "ft_kernels.hpp"

namespace ftsgemm {

// balanced-range inverse map: the workgroup owning work unit u, for a
// grid of G ranges over `total` units (q = total/G, rem = total%G)
__device__ inline int sk_wg_of_unit(int u, int q, int rem) {
  return (u < rem * (q + 1)) ? u / (q + 1)
                             : rem + (u - rem * (q + 1)) / q;
}

// Split-tile combine pass: one workgroup per output tile, with the SAME
// thread geometry as the main kernel so thread t reads back exactly the
// fragment elements thread t of each contributor wrote.  Fully-owned
// tiles exit immediately (the main kernel already wrote them).
template <int BM, int BN, int WM, int WN, int MM>
static __global__ __launch_bounds__(64 * (BM / WM) * (BN / WN))
void sk_fixup_kernel(int M, int N, int upt, int G, float alpha, float beta,
                     float* __restrict__ C,
                     const float* __restrict__ partials) {
  using T = mfma_traits<MM>;
  constexpr int NREG = T::nreg;
  constexpr int WAVES_N = BN / WN;
  constexpr int FM = WM / MM, FN = WN / MM;
  constexpr int TPT = FM * FN * NREG;
  const int tile = blockIdx.x;
  const int ntm = M / BM;
  const int total = ntm * (N / BN) * upt;
  const int q = total / G, rem = total % G;
  const int g0 = sk_wg_of_unit(tile * upt, q, rem);
  const int gl = sk_wg_of_unit(tile * upt + upt - 1, q, rem);
  if (g0 == gl) return;  // fully owned -> already epilogued

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane / MM;
  const int r = lane % MM;
  const int wm_idx = wave / WAVES_N, wn_idx = wave % WAVES_N;
  const int wi0 = wm_idx * WM;
  const int wj0 = wn_idx * WN;
  const int bx = tile % ntm, by = tile / ntm;
  const int im0 = bx * BM, jn0 = by * BN;

  float s[TPT];
  {  // g0's tail partial (slot 2*g0+1), then every head partial in (g0,gl]
    const float* slot = partials + (size_t)(2 * g0 + 1) * (BM * BN) +
                        tid * TPT;
#pragma unroll
    for (int e = 0; e < TPT; ++e) s[e] = slot[e];
  }
  for (int gc = g0 + 1; gc <= gl; ++gc) {
    const float* slot = partials + (size_t)(2 * gc) * (BM * BN) + tid * TPT;
#pragma unroll
    for (int e4 = 0; e4 < TPT; e4 += 4) {
      const f32x4 v = *(const f32x4*)(slot + e4);
#pragma unroll
      for (int q4 = 0; q4 < 4; ++q4) s[e4 + q4] += v[q4];
    }
  }
#pragma unroll
  for (int fm = 0; fm < FM; ++fm)
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      const int j = jn0 + wj0 + fn * MM + r;
      float* colbase = C + (size_t)j * M + im0 + wi0 + fm * MM;
#pragma unroll
      for (int g4 = 0; g4 < NREG / 4; ++g4) {
        float* p4 = colbase + 4 * sub + 8 * g4;
        const int e = (fm * FN + fn) * NREG + 4 * g4;
        f32x4 out;
        if (beta != 0.f) {
          const f32x4 prev = *(const f32x4*)p4;
#pragma unroll
          for (int q4 = 0; q4 < 4; ++q4)
            out[q4] = alpha * s[e + q4] + beta * prev[q4];
        } else {
#pragma unroll
          for (int q4 = 0; q4 < 4; ++q4) out[q4] = alpha * s[e + q4];
        }
        *(f32x4*)p4 = out;
      }
    }
}

template <int BM, int BN, int BK, int WM, int WN, int MM, bool ABFT,
          bool INJECT, int OCC = 2>
__global__ __launch_bounds__(64 * (BM / WM) * (BN / WN), OCC)
void sgemm_mfma_streamk(int M, int N, int K, const float* __restrict__ A,
                        const float* __restrict__ B, float* __restrict__ C,
                        float alpha, float beta, int istride, float tau,
                        float inj_mag, const float* __restrict__ SA, int sstr,
                        float* __restrict__ partials) {
  using T = mfma_traits<MM>;
  constexpr int KSTEP = T::kstep;
  constexpr int NREG = T::nreg;
  constexpr int WAVES_M = BM / WM, WAVES_N = BN / WN;
  constexpr int NWAVES = WAVES_M * WAVES_N;
  constexpr int THREADS = NWAVES * 64;
  constexpr int FM = WM / MM, FN = WN / MM;
  constexpr int BUF = (BM + BN) * BK;
  constexpr int PPS = 64 / BK;  // panels per strip window / work unit
  static_assert(64 % BK == 0 && BK <= 64, "stream-K unit is 64 k");
  constexpr int STRIP_OFF = 2 * BUF;
  constexpr int LDS_FLOATS = ABFT ? (STRIP_OFF + NWAVES * 256) : (2 * BUF);

  __shared__ __attribute__((aligned(16))) float lds[LDS_FLOATS];

  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane / MM;
  const int r = lane % MM;
  const int wm_idx = wave / WAVES_N, wn_idx = wave % WAVES_N;
  const int wi0 = wm_idx * WM;
  const int wj0 = wn_idx * WN;

  const int ntm = M / BM;
  const int upt = K >> 6;  // units (64-k windows) per tile
  const int total = ntm * (N / BN) * upt;
  // balanced contiguous unit ranges: first (total % G) workgroups take one
  // extra unit
  const int g = blockIdx.x;
  const int q = total / gridDim.x, rem = total % gridDim.x;
  int u = g * q + (g < rem ? g : rem);
  const int u_end = u + q + (g < rem ? 1 : 0);

  constexpr int GA = (BM * BK) / (THREADS * 4);
  constexpr int GB = (BN * BK) / (THREADS * 4);
  static_assert(GA >= 1 && GB >= 1, "tile too small for this thread count");

  auto stage = [&](int qb, int k0, int im0, int jn0)
      __attribute__((always_inline)) {
    float* dstA = &lds[qb * BUF];
    float* dstB = &lds[qb * BUF + BM * BK];
#pragma unroll
    for (int t = 0; t < GA; ++t) {
      const int f = (t * THREADS + tid) * 4;
      const int k = f / BM, i = f % BM;
      const float* gp = A + (im0 + i) + (size_t)(k0 + k) * M;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gp,
          (__attribute__((address_space(3))) void*)(dstA +
                                                    (t * THREADS + wave * 64) *
                                                        4),
          16, 0, 0);
    }
#pragma unroll
    for (int t = 0; t < GB; ++t) {
      const int f = (t * THREADS + tid) * 4;
      const int k = f / BN, j = f % BN;
      const float* gp = B + (jn0 + j) + (size_t)(k0 + k) * N;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gp,
          (__attribute__((address_space(3))) void*)(dstB +
                                                    (t * THREADS + wave * 64) *
                                                        4),
          16, 0, 0);
    }
  };

  // interleaved (plain, weighted) segment-sum pairs: see ft_kernels.hpp
  auto strip_stage = [&](int pb, int k0, int segA)
      __attribute__((always_inline)) {
    const float* ga = SA + (size_t)segA * 2 * sstr + 2 * k0 + lane;
    float* dst = &lds[STRIP_OFF + wave * 256 + pb * 128];
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)ga,
        (__attribute__((address_space(3))) void*)dst, 4, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)(ga + 64),
        (__attribute__((address_space(3))) void*)(dst + 64), 4, 0, 0);
  };

  // Accumulators and the ABFT verify machinery are declared ONCE, outside
  // the tile loop, and re-zeroed per segment — declaring them inside the
  // loop body made the allocator keep per-iteration copies alive and spill
  // 142 VGPRs in the fused variants (measured; the plain variant fit).
  typename T::acc_t acc[FM][FN];
  float cc[FN];
  float cw[FN];
#pragma unroll
  for (int fm = 0; fm < FM; ++fm)
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) acc[fm][fn] = {};
#pragma unroll
  for (int fn = 0; fn < FN; ++fn) cc[fn] = cw[fn] = 0.f;

    // verify/locate/correct: identical maths to the classic kernel
    // (csrc/ft_kernels.hpp locate_correct/verify_correct), operating on
    // this workgroup's PARTIAL k-range accumulation.
    auto locate_correct = [&]() __attribute__((always_inline)) {
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        // fp64 plain sum (correction magnitude), fp32 weighted sum (row
        // index only) — see the classic kernel's locate_correct note.
        int sub_o = sub;  // fresh opaque copy per fn (see classic)
        asm volatile("" : "+v"(sub_o));
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

  while (u < u_end) {
    const int tile = u / upt;
    const int w_lo = u - tile * upt;
    const int seg_units = (u_end - u < upt - w_lo) ? (u_end - u)
                                                   : (upt - w_lo);
    const int w_hi = w_lo + seg_units;
    const int bx = tile % ntm, by = tile / ntm;
    const int im0 = bx * BM, jn0 = by * BN;
    const int segA = bx * WAVES_M + wm_idx;
    const int kbase = w_lo << 6;
    const int npan = seg_units * PPS;

    stage(0, kbase, im0, jn0);
    if constexpr (ABFT) strip_stage(w_lo & 1, kbase, segA);
    FT_PARA_SYNC();
    __syncthreads();

    const int PPG = istride * PPS;  // panels per inject+verify group
    int p = 0;
    while (p < npan) {
      if constexpr (INJECT) {
        // deterministic rotating victim per (tile, window-group)
        if (tid == ((unsigned)((w_lo + p / PPS) * 67u + tile * 13u) %
                    THREADS))
          acc[0][0][0] += inj_mag;
      }
      const int burst_end = (p + PPG < npan) ? p + PPG : npan;
      for (; p < burst_end; ++p) {
        const int qb = p & 1;
        if (p + 1 < npan) {
          stage(qb ^ 1, kbase + (p + 1) * BK, im0, jn0);
          if constexpr (ABFT) {
            if ((p + 1) % PPS == 0)
              strip_stage((w_lo + (p + 1) / PPS) & 1,
                          kbase + (p + 1) * BK, segA);
          }
        }
        FT_PARA_SYNC();
        const float* As = &lds[qb * BUF];
        const float* Bs = &lds[qb * BUF + BM * BK];
        const float* strip =
            ABFT ? &lds[STRIP_OFF + wave * 256 + ((w_lo + p / PPS) & 1) * 128 +
                        (p % PPS) * BK * 2]
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
            const f32x2 sw = *(const f32x2*)(strip + 2 * kloc);
#pragma unroll
            for (int fn = 0; fn < FN; ++fn) {
              cc[fn] = fmaf(sw[0], b[fn], cc[fn]);
              cw[fn] = fmaf(sw[1], b[fn], cw[fn]);
            }
          }
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
    }

    // ---- tile contribution ----
    // A tile fully inside this range takes the direct alpha/beta epilogue.
    // Split-tile segments write their raw partial to a private slot (head
    // partial, w_lo>0 -> slot 2g; tail partial, w_lo==0 && w_hi<upt ->
    // slot 2g+1); the fixup kernel launched AFTER this one combines them.
    // No flags, no spin, no co-residency assumption: an earlier protocol
    // had split-tile owners acquire-spin on contributor flags, which
    // deadlocked whenever the dispatcher did not make the whole grid
    // co-resident (reproduced: fresh-process CLI runs dispatched
    // partially and hung at N=4608; the same binary was fine once the
    // process had prior GPU activity).  Kernel-boundary ordering makes
    // the partials visible to the fixup kernel with no fences at all.
    constexpr int TPT = FM * FN * NREG;  // per-thread floats of one tile
    const bool full_seg = (w_lo == 0) && (w_hi == upt);
    if (!full_seg) {
      float* slot = partials +
                    ((size_t)(2 * g + (w_lo == 0 ? 1 : 0))) * (BM * BN) +
                    tid * TPT;
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int fn = 0; fn < FN; ++fn)
#pragma unroll
          for (int g4 = 0; g4 < NREG / 4; ++g4) {
            f32x4 v;
#pragma unroll
            for (int q4 = 0; q4 < 4; ++q4) v[q4] = acc[fm][fn][4 * g4 + q4];
            *(f32x4*)(slot + (fm * FN + fn) * NREG + 4 * g4) = v;
          }
    } else {
      // direct alpha/beta epilogue (identical to the classic kernel)
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int fn = 0; fn < FN; ++fn) {
          const int j = jn0 + wj0 + fn * MM + r;
          float* colbase = C + (size_t)j * M + im0 + wi0 + fm * MM;
#pragma unroll
          for (int g4 = 0; g4 < NREG / 4; ++g4) {
            float* p4 = colbase + 4 * sub + 8 * g4;
            f32x4 out;
            if (beta != 0.f) {
              const f32x4 prev = *(const f32x4*)p4;
#pragma unroll
              for (int q4 = 0; q4 < 4; ++q4)
                out[q4] = alpha * acc[fm][fn][4 * g4 + q4] + beta * prev[q4];
            } else {
#pragma unroll
              for (int q4 = 0; q4 < 4; ++q4)
                out[q4] = alpha * acc[fm][fn][4 * g4 + q4];
            }
            *(f32x4*)p4 = out;
          }
        }
    }
    u += seg_units;
    if (u < u_end) {  // re-zero for the next tile segment
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int fn = 0; fn < FN; ++fn) acc[fm][fn] = {};
      if constexpr (ABFT) {
#pragma unroll
        for (int fn = 0; fn < FN; ++fn) cc[fn] = cw[fn] = 0.f;
      }
    }
  }
}

}  // namespace ftsgemm
