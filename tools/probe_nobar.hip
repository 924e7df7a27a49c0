// probe_nobar.hip — BARRIER-FREE SGEMM: every wave stages its own private
// LDS copy of its operands (its 64-row A slice + a duplicate of the whole
// 128-col B panel), so no __syncthreads exists anywhere and each wave
// self-paces its glds pipeline with counted per-wave s_waitcnt.  Trades
// 4x B staging traffic (L2/L3-served) for zero barrier parks — the
// measured 13.2% parked-wave cost of the shared-LDS design.
//
// Tile: 256x128xBK as 4 waves of 64x128 (FM=2, FN=4, mfma_f32_32x32x2).
// LDS: per wave 2 x (64+128) x BK floats; BK=16 -> 96 KB (1 block/CU),
// BK=8 -> 48 KB (2 blocks/CU).
//
// Build: hipcc -x hip --offload-arch=gfx950 -O3 tools/probe_nobar.hip -o bin/probe_nobar

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;

#define BM 256
#define BN 128

template <int BK, int OCC>
__global__ __launch_bounds__(256, OCC) void k_nobar(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  constexpr int WBUF = (64 + BN) * BK;  // floats per wave per buffer
  __shared__ __attribute__((aligned(16))) float lds[4 * 2 * WBUF];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * BM, jn0 = blockIdx.y * BN;

  float* W = &lds[wave * 2 * WBUF];  // this wave's private region

  f32x16 acc[2][4] = {};

  constexpr int GA = 64 * BK / 256;   // glds per A slice (16 B x 64 lanes)
  constexpr int GB = BN * BK / 256;   // glds per B panel copy
  auto stage = [&](int q, int k0) __attribute__((always_inline)) {
    float* wa = W + q * WBUF;
    float* wb = wa + 64 * BK;
#pragma unroll
    for (int t = 0; t < GA; ++t) {
      const int f = (t * 64 + lane) * 4;
      const int k = f / 64, i = f % 64;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(A + (im0 + wi0 + i) +
                                                          (size_t)(k0 + k) * M),
          (__attribute__((address_space(3))) void*)(wa + t * 256), 16, 0, 0);
    }
#pragma unroll
    for (int t = 0; t < GB; ++t) {
      const int f = (t * 64 + lane) * 4;
      const int k = f / BN, j = f % BN;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)(B + (jn0 + j) +
                                                          (size_t)(k0 + k) * N),
          (__attribute__((address_space(3))) void*)(wb + t * 256), 16, 0, 0);
    }
  };
  constexpr int LD = GA + GB;

  const int niter = K / BK;
  stage(0, 0);
  for (int it = 0; it < niter; ++it) {
    const int q = it & 1;
    if (it + 1 < niter) {
      stage(q ^ 1, (it + 1) * BK);
      asm volatile("s_waitcnt vmcnt(%0)" ::"n"(LD) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    const float* Wa = W + q * WBUF;
    const float* Wb = Wa + 64 * BK;
#pragma unroll
    for (int kk = 0; kk < BK / 2; ++kk) {
      const int kloc = kk * 2 + sub;
      float a[2], b[4];
      a[0] = Wa[kloc * 64 + r];
      a[1] = Wa[kloc * 64 + 32 + r];
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) b[fn] = Wb[kloc * BN + fn * 32 + r];
      __builtin_amdgcn_iglp_opt(0);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              a[fm], b[fn], acc[fm][fn], 0, 0, 0);
    }
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
      const int j = jn0 + fn * 32 + r;
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

__global__ void fill_lcg(float* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  unsigned s = seed ^ (unsigned)(i * 2654435761u);
  s ^= s << 13; s ^= s >> 17; s ^= s << 5;
  p[i] = ((s >> 8) * (1.0f / 16777216.0f)) * 1.8f - 0.9f;
}

__global__ void max_diff(const float* x, const float* y, size_t n,
                         float* out) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  float d = 0.f;
  if (i < n) d = fabsf(x[i] - y[i]);
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) d = fmaxf(d, __shfl_xor(d, m, 64));
  if ((threadIdx.x & 63) == 0 && d > *out)
    atomicMax((int*)out, __float_as_int(d));
}

__global__ void k_ref(int n, const float* A, const float* B, float* C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int j = blockIdx.y;
  if (i >= n) return;
  float s = 0.f;
  for (int k = 0; k < n; ++k) s += A[i + (size_t)k * n] * B[j + (size_t)k * n];
  C[i + (size_t)j * n] = s;
}

template <int BK, int OCC>
static void run(const char* name, int n, const float* dA, const float* dB,
                float* dC, const float* dRef, float* dMax, int reps) {
  dim3 grid(n / BM, n / BN), block(256);
  hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
  hipLaunchKernelGGL((k_nobar<BK, OCC>), grid, block, 0, 0, n, n, n, dA, dB,
                     dC, 1.f, 0.f);
  float md = 0.f;
  hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
  hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256), dim3(256),
                     0, 0, dC, dRef, (size_t)n * n, dMax);
  hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
  hipEvent_t b0, b1;
  hipEventCreate(&b0);
  hipEventCreate(&b1);
  for (int w = 0; w < 2; ++w)
    hipLaunchKernelGGL((k_nobar<BK, OCC>), grid, block, 0, 0, n, n, n, dA,
                       dB, dC, 1.f, -1.5f);
  hipDeviceSynchronize();
  hipEventRecord(b0);
  for (int rr = 0; rr < reps; ++rr)
    hipLaunchKernelGGL((k_nobar<BK, OCC>), grid, block, 0, 0, n, n, n, dA,
                       dB, dC, 1.f, -1.5f);
  hipEventRecord(b1);
  hipEventSynchronize(b1);
  float ms;
  hipEventElapsedTime(&ms, b0, b1);
  printf("N=%d %-28s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n, name,
         2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
         hipGetErrorString(hipGetLastError()));
  fflush(stdout);
}

int main(int argc, char** argv) {
  setbuf(stdout, NULL);
  int n = argc > 1 ? atoi(argv[1]) : 4096;
  size_t nn = (size_t)n * n, bytes = nn * 4;
  float *dA, *dB, *dC, *dRef, *dMax;
  hipMalloc(&dA, bytes);
  hipMalloc(&dB, bytes);
  hipMalloc(&dC, bytes);
  hipMalloc(&dRef, bytes);
  hipMalloc(&dMax, 4);
  hipLaunchKernelGGL(fill_lcg, dim3((nn + 255) / 256), dim3(256), 0, 0, dA,
                     nn, 1u);
  hipLaunchKernelGGL(fill_lcg, dim3((nn + 255) / 256), dim3(256), 0, 0, dB,
                     nn, 2u);
  hipLaunchKernelGGL(k_ref, dim3((n + 255) / 256, n), dim3(256), 0, 0, n, dA,
                     dB, dRef);
  hipDeviceSynchronize();
  for (int round = 0; round < 3; ++round) {
    printf("--- round %d\n", round);
    run<16, 1>("N1 nobar 256x128x16 1blk", n, dA, dB, dC, dRef, dMax, 10);
    run<8, 2>("N2 nobar 256x128x8  2blk", n, dA, dB, dC, dRef, dMax, 10);
    run<8, 1>("N3 nobar 256x128x8  occ1", n, dA, dB, dC, dRef, dMax, 10);
  }
  return 0;
}
