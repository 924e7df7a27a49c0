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
// Design (work-centric, CUTLASS-stream-K-like but simplified around the
// ABFT strip windows): the GEMM is a list of `total = ntiles * (K/64)`
// work units — one unit = one 64-k strip window of one output tile (4
// BK=16 LDS panels, exactly the granularity the ABFT checksum strips are
// staged at).  A fixed grid of G workgroups (G = CUs x occupancy, never
// more than total) each processes a CONTIGUOUS unit range, k-fastest:
// perfect load balance at any size, no tail round.  A tile fully owned by
// one workgroup takes the normal alpha/beta epilogue; a tile split across
// workgroups is combined with native f32 global atomics
// (global_atomic_add_f32 via unsafeAtomicAdd) after a one-pass C prescale
// kernel has applied beta — fp32 add-order nondeterminism across splits is
// far below the 1e-2/1e-2 verify tolerance at the reference operand scale.
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

// C *= beta in one grid-stride f32x4 pass (beta==0 zero-fills: stream-K
// split tiles ACCUMULATE into C, so beta must be applied up front exactly
// once — the classic kernels instead fold beta into their epilogue).
// static: per-TU copy (only the stream-K TUs instantiate it); a plain
// external __global__ here would collide at link across kernel_*_sk.hip.
static __global__ __launch_bounds__(256) void prescale_kernel(
    size_t total4, float beta, float* __restrict__ C) {
  const size_t stride = (size_t)gridDim.x * 256;
  for (size_t i = (size_t)blockIdx.x * 256 + threadIdx.x; i < total4;
       i += stride) {
    f32x4* p = (f32x4*)C + i;
    if (beta == 0.f) {
      *p = f32x4{0.f, 0.f, 0.f, 0.f};
    } else {
      f32x4 v = *p;
#pragma unroll
      for (int u = 0; u < 4; ++u) v[u] *= beta;
      *p = v;
    }
  }
}

template <int BM, int BN, int BK, int WM, int WN, int MM, bool ABFT,
          bool INJECT, int OCC = 2>
__global__ __launch_bounds__(64 * (BM / WM) * (BN / WN), OCC)
void sgemm_mfma_streamk(int M, int N, int K, const float* __restrict__ A,
                        const float* __restrict__ B, float* __restrict__ C,
                        float alpha, int istride, float tau, float inj_mag,
                        const float* __restrict__ SA, int sstr) {
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

  const int segsA = ABFT ? (M / WM) : 0;
  auto strip_stage = [&](int pb, int k0, int segA)
      __attribute__((always_inline)) {
    const float* ga = SA + (size_t)segA * sstr + k0 + lane;
    const float* gw = SA + (size_t)(segsA + segA) * sstr + k0 + lane;
    float* dst = &lds[STRIP_OFF + wave * 256 + pb * 128];
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)ga,
        (__attribute__((address_space(3))) void*)dst, 4, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gw,
        (__attribute__((address_space(3))) void*)(dst + 64), 4, 0, 0);
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

    typename T::acc_t acc[FM][FN] = {};
    float cc[FN] = {};
    float cw[FN] = {};

    // verify/locate/correct: identical maths to the classic kernel
    // (csrc/ft_kernels.hpp locate_correct/verify_correct), operating on
    // this workgroup's PARTIAL k-range accumulation.
    auto locate_correct = [&]() __attribute__((always_inline)) {
      int sub_o = sub;
      asm volatile("" : "+v"(sub_o));
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        double colp = 0.0, colw = 0.0;
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
#pragma unroll
          for (int reg = 0; reg < NREG; ++reg) {
            const double v = (double)acc[fm][fn][reg];
            colp += v;
            colw = fma((double)(fm * MM + acc_row(reg, sub_o)), v, colw);
          }
        const float rc =
            (float)(dslice_sum<MM>(colp) - (double)slice_sum<MM>(cc[fn]));
        const float rw =
            (float)(dslice_sum<MM>(colw) - (double)slice_sum<MM>(cw[fn]));
        const bool cbad = fabsf(rc) > tau;
        const int row = (int)rintf(rw / (cbad ? rc : 1.f));
#pragma unroll
        for (int fm = 0; fm < FM; ++fm)
#pragma unroll
          for (int reg = 0; reg < NREG; ++reg) {
            const bool hit = cbad && (fm * MM + acc_row(reg, sub_o) == row);
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

    stage(0, kbase, im0, jn0);
    if constexpr (ABFT) strip_stage(w_lo & 1, kbase, segA);
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
        const float* As = &lds[qb * BUF];
        const float* Bs = &lds[qb * BUF + BM * BK];
        const float* strip =
            ABFT ? &lds[STRIP_OFF + wave * 256 + ((w_lo + p / PPS) & 1) * 128 +
                        (p % PPS) * BK]
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
            const float sa = strip[kloc];
            const float saw = strip[64 + kloc];
#pragma unroll
            for (int fn = 0; fn < FN; ++fn) {
              cc[fn] = fmaf(sa, b[fn], cc[fn]);
              cw[fn] = fmaf(saw, b[fn], cw[fn]);
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
    // full owner: C was pre-scaled by beta, sole writer -> plain RMW;
    // split tile: accumulate with native f32 global atomics.
    const bool full = (w_lo == 0) && (w_hi == upt);
#pragma unroll
    for (int fm = 0; fm < FM; ++fm)
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        const int j = jn0 + wj0 + fn * MM + r;
        float* colbase = C + (size_t)j * M + im0 + wi0 + fm * MM;
#pragma unroll
        for (int g4 = 0; g4 < NREG / 4; ++g4) {
          float* p4 = colbase + 4 * sub + 8 * g4;
          if (full) {
            const f32x4 prev = *(const f32x4*)p4;
            f32x4 out;
#pragma unroll
            for (int v = 0; v < 4; ++v)
              out[v] = fmaf(alpha, acc[fm][fn][4 * g4 + v], prev[v]);
            *(f32x4*)p4 = out;
          } else {
#pragma unroll
            for (int v = 0; v < 4; ++v)
              unsafeAtomicAdd(p4 + v, alpha * acc[fm][fn][4 * g4 + v]);
          }
        }
      }
    u += seg_units;
  }
}

}  // namespace ftsgemm
