// probe_ablate.hip — ablation probe for the fused-ABFT huge kernel.
// Variants co-timed in one process (within-probe A/B):
//   V0 plain            : no ABFT
//   V1 abft-full (vw1)  : sums pass + extra barrier + encode + 1 verify
//   V2 sums+barrier only: no encode fmas, no verify
//   V3 encode only      : no sums pass/barrier (sa/sb from LDS, stale)
//   V4 fma only         : encode with constant sa/sb (no LDS reads)
// Build: hipcc -x hip --offload-arch=gfx950 -O3 tools/probe_ablate.hip -o bin/probe_ablate
// NOT part of the shipped library (numerics of V2..V4 are meaningless).

#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;

#define BM 128
#define BN 128
#define BK 32
#define WM 64
#define WN 64

template <bool SUMS, bool EXTRA_BARRIER, bool ENC_LDS, bool ENC_FMA,
          bool VERIFY>
__global__ __launch_bounds__(256) void k_ablate(int M, int N, int K,
                                                const float* __restrict__ A,
                                                const float* __restrict__ B,
                                                float* __restrict__ C,
                                                float alpha, float beta) {
  constexpr int BUF = (BM + BN) * BK;
  constexpr int SA_OFF = 2 * BUF, SB_OFF = SA_OFF + 2 * BK;
  __shared__ __attribute__((aligned(16))) float lds[SB_OFF + 2 * BK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wm_idx = wave >> 1, wn_idx = wave & 1;
  const int wi0 = wm_idx * WM, wj0 = wn_idx * WN;
  const int im0 = blockIdx.x * BM, jn0 = blockIdx.y * BN;
  f32x16 acc[2][2] = {};
  float cr[2] = {}, cc[2] = {};

  auto stage = [&](int q, int k0) {
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int f = (t * 256 + tid) * 4;
      const int k = f / BM, i = f % BM;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(A + (im0 + i) +
                                                          (size_t)(k0 + k) * M),
          (__attribute__((address_space(3))) void*)(&lds[q * BUF] +
                                                    (t * 256 + wave * 64) * 4),
          16, 0, 0);
    }
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int f = (t * 256 + tid) * 4;
      const int k = f / BN, j = f % BN;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(B + (jn0 + j) +
                                                          (size_t)(k0 + k) * N),
          (__attribute__((address_space(3))) void*)(&lds[q * BUF + BM * BK] +
                                                    (t * 256 + wave * 64) * 4),
          16, 0, 0);
    }
  };

  auto panel_sums = [&](int q) {
    const float* As = &lds[q * BUF];
    const float* Bs = &lds[q * BUF + BM * BK];
    {
      const int task = tid >> 2, st = tid & 3;  // 64 tasks x 4 threads
      const int wmi = task / BK, k = task % BK;
      const float* src = As + k * BM + wmi * WM + st * 16;
      float s = 0.f;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const f32x4 v = *(const f32x4*)(src + 4 * u);
        s += (v[0] + v[1]) + (v[2] + v[3]);
      }
      s += __shfl_xor(s, 1, 64);
      s += __shfl_xor(s, 2, 64);
      if (st == 0) lds[SA_OFF + task] = s;
    }
    {
      const int task = tid >> 2, st = tid & 3;
      const int wni = task / BK, k = task % BK;
      const float* src = Bs + k * BN + wni * WN + st * 16;
      float s = 0.f;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const f32x4 v = *(const f32x4*)(src + 4 * u);
        s += (v[0] + v[1]) + (v[2] + v[3]);
      }
      s += __shfl_xor(s, 1, 64);
      s += __shfl_xor(s, 2, 64);
      if (st == 0) lds[SB_OFF + task] = s;
    }
  };

  stage(0, 0);
  __syncthreads();
  const int niter = K / BK;
  for (int it = 0; it < niter; ++it) {
    const int q = it & 1;
    if constexpr (SUMS) panel_sums(q);
    if constexpr (EXTRA_BARRIER) __syncthreads();
    if (it + 1 < niter) stage(q ^ 1, (it + 1) * BK);
    const float* As = &lds[q * BUF];
    const float* Bs = &lds[q * BUF + BM * BK];
#pragma unroll
    for (int kk = 0; kk < BK / 2; ++kk) {
      const int kloc = kk * 2 + sub;
      float a[2], b[2];
      a[0] = As[kloc * BM + wi0 + r];
      a[1] = As[kloc * BM + wi0 + 32 + r];
      b[0] = Bs[kloc * BN + wj0 + r];
      b[1] = Bs[kloc * BN + wj0 + 32 + r];
      if constexpr (ENC_FMA) {
        float sa = 1.f, sb = 1.f;
        if constexpr (ENC_LDS) {
          sa = lds[SA_OFF + wm_idx * BK + kloc];
          sb = lds[SB_OFF + wn_idx * BK + kloc];
        }
        cr[0] = fmaf(a[0], sb, cr[0]);
        cr[1] = fmaf(a[1], sb, cr[1]);
        cc[0] = fmaf(sa, b[0], cc[0]);
        cc[1] = fmaf(sa, b[1], cc[1]);
      }
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              a[fm], b[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
  }

  if constexpr (VERIFY) {
    // single cheap detect: total-sum compare (keeps cr/cc live)
    float tot = 0.f;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int g = 0; g < 16; ++g) tot += acc[fm][fn][g];
    float chk = cr[0] + cr[1];
#pragma unroll
    for (int m = 1; m < 64; m <<= 1) {
      tot += __shfl_xor(tot, m, 64);
      chk += __shfl_xor(chk, m, 64);
    }
    if (fabsf(tot - chk) > 1e30f) acc[0][0][0] += 1.f;  // never taken
  } else {
    asm volatile("" ::"v"(cr[0]), "v"(cr[1]), "v"(cc[0]), "v"(cc[1]));
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int j = jn0 + wj0 + fn * 32 + r;
      float* colbase = C + (size_t)j * M + im0 + wi0 + fm * 32;
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        float* p = colbase + 4 * sub + 8 * g;
        const f32x4 prev = *(const f32x4*)p;
        f32x4 out;
#pragma unroll
        for (int u = 0; u < 4; ++u)
          out[u] = alpha * acc[fm][fn][4 * g + u] + beta * prev[u];
        *(f32x4*)p = out;
      }
    }
}

#define HIP_CALL(x)                                                     \
  do {                                                                  \
    hipError_t e_ = (x);                                                \
    if (e_ != hipSuccess) {                                             \
      printf("HIP err %s @%d\n", hipGetErrorString(e_), __LINE__);      \
      return 1;                                                         \
    }                                                                   \
  } while (0)

template <bool S, bool XB, bool EL, bool EF, bool V>
static float time_variant(const char* name, int n, const float* dA,
                          const float* dB, float* dC, int reps) {
  dim3 grid(n / BM, n / BN), block(256);
  hipEvent_t b0, b1;
  hipEventCreate(&b0);
  hipEventCreate(&b1);
  for (int w = 0; w < 2; ++w)
    hipLaunchKernelGGL((k_ablate<S, XB, EL, EF, V>), grid, block, 0, 0, n, n,
                       n, dA, dB, dC, 1.f, -1.5f);
  hipDeviceSynchronize();
  hipEventRecord(b0);
  for (int r = 0; r < reps; ++r)
    hipLaunchKernelGGL((k_ablate<S, XB, EL, EF, V>), grid, block, 0, 0, n, n,
                       n, dA, dB, dC, 1.f, -1.5f);
  hipEventRecord(b1);
  hipEventSynchronize(b1);
  float ms;
  hipEventElapsedTime(&ms, b0, b1);
  double gf = 2.0 * n * n * n * reps / (ms * 1e-3) / 1e9;
  printf("%-28s %8.0f GFLOPS\n", name, gf);
  return ms;
}

int main() {
  const int n = 4096, reps = 10;
  size_t bytes = (size_t)n * n * 4;
  float *dA, *dB, *dC;
  HIP_CALL(hipMalloc(&dA, bytes));
  HIP_CALL(hipMalloc(&dB, bytes));
  HIP_CALL(hipMalloc(&dC, bytes));
  HIP_CALL(hipMemset(dA, 0x3c, bytes));
  HIP_CALL(hipMemset(dB, 0x3c, bytes));
  HIP_CALL(hipMemset(dC, 0, bytes));
  for (int round = 0; round < 3; ++round) {
    printf("--- round %d\n", round);
    time_variant<false, false, false, false, false>("V0 plain", n, dA, dB,
                                                    dC, reps);
    time_variant<true, true, true, true, true>("V1 abft-full-vw1", n, dA, dB,
                                               dC, reps);
    time_variant<true, true, false, false, false>("V2 sums+barrier", n, dA,
                                                  dB, dC, reps);
    time_variant<false, false, true, true, false>("V3 encode(lds)", n, dA,
                                                  dB, dC, reps);
    time_variant<false, false, false, true, false>("V4 encode(const)", n, dA,
                                                   dB, dC, reps);
    time_variant<true, false, true, true, false>("V5 sums-nobarrier+enc", n,
                                                 dA, dB, dC, reps);
  }
  return 0;
}
