// probe_dtva.hip — "direct-to-VGPR A" SGEMM structure (what rocBLAS/Tensile
// uses on gfx950 for SGEMM: Cijk_..._MT256x128x8_MI16x16x4_..._DTVA1_LPB16,
// LDS 4.6 KB, VGPR 128): waves partition M only, so each wave's A-slice is
// private and loads straight to registers (no LDS, no cross-wave
// dependency); only the shared B panel goes through LDS, staged
// register->ds_write with the write placed between two barriers (Tensile
// 1LDSB / guide T14 form).  The panel seam exposes only ~one ds_read
// latency instead of a cross-wave DMA drain.
//
// Tile: 256x128xBK, 4 waves as 4(M)x1(N): WM=64, WN=128, FM=2, FN=4,
// v_mfma_f32_32x32x2_f32.
//
// Build: hipcc -x hip --offload-arch=gfx950 -O3 tools/probe_dtva.hip -o bin/probe_dtva

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;

#define BM 256
#define BN 128

template <int BK, bool DBUF>
__global__ __launch_bounds__(256, 2) void k_dtva(int M, int N, int K,
                                                 const float* __restrict__ A,
                                                 const float* __restrict__ B,
                                                 float* __restrict__ C,
                                                 float alpha, float beta) {
  constexpr int NB = DBUF ? 2 : 1;
  __shared__ __attribute__((aligned(16))) float Bs[NB * BN * BK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * BM, jn0 = blockIdx.y * BN;

  f32x16 acc[2][4] = {};
  float areg[2][2][BK / 2];  // [buf][fm][kk]
  f32x4 breg[BN * BK / 1024];  // per-thread B staging (j-contiguous 16 B)

  constexpr int BCH = BN * BK / 1024;  // f32x4 chunks per thread
  auto load_a = [&](int buf, int k0) __attribute__((always_inline)) {
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int kk = 0; kk < BK / 2; ++kk)
        areg[buf][fm][kk] =
            A[im0 + wi0 + fm * 32 + r + (size_t)(k0 + kk * 2 + sub) * M];
  };
  auto load_b = [&](int k0) __attribute__((always_inline)) {
#pragma unroll
    for (int u = 0; u < BCH; ++u) {
      const int f = (u * 256 + tid) * 4;
      const int k = f / BN, j = f % BN;
      breg[u] = *(const f32x4*)(B + (jn0 + j) + (size_t)(k0 + k) * N);
    }
  };
  auto write_b = [&](int q) __attribute__((always_inline)) {
#pragma unroll
    for (int u = 0; u < BCH; ++u)
      *(f32x4*)(&Bs[q * BN * BK] + (u * 256 + tid) * 4) = breg[u];
  };

  const int niter = K / BK;
  load_a(0, 0);
  load_b(0);
  for (int it = 0; it < niter; ++it) {
    const int q = DBUF ? (it & 1) : 0;
    __syncthreads();  // readers of Bs[q] (panel it-NB) done
    write_b(q);       // panel it (waits vmcnt for breg internally)
    if (it + 1 < niter) load_b((it + 1) * BK);
    __syncthreads();  // Bs[q] visible (lgkm drained inside)
    if (it + 1 < niter) load_a((it & 1) ^ 1, (it + 1) * BK);
    const float* Bp = &Bs[q * BN * BK];
    const int ab = it & 1;
#pragma unroll
    for (int kk = 0; kk < BK / 2; ++kk) {
      const int kloc = kk * 2 + sub;
      float b[4];
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) b[fn] = Bp[kloc * BN + fn * 32 + r];
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              areg[ab][fm][kk], b[fn], acc[fm][fn], 0, 0, 0);
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

// D6: single barrier per panel — B double-buffered in LDS, the ds_write of
// panel it+1 (from registers loaded during it-1) overlaps the MFMA loop of
// panel it; A direct-to-VGPR one panel ahead.  Tensile-like PGR2/PLR.
template <int BK>
__global__ __launch_bounds__(256, 2) void k_dtva2(int M, int N, int K,
                                                  const float* __restrict__ A,
                                                  const float* __restrict__ B,
                                                  float* __restrict__ C,
                                                  float alpha, float beta) {
  __shared__ __attribute__((aligned(16))) float Bs[2 * BN * BK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * BM, jn0 = blockIdx.y * BN;

  f32x16 acc[2][4] = {};
  float areg[2][2][BK / 2];
  constexpr int BCH = BN * BK / 1024;
  f32x4 breg[BCH];

  auto load_a = [&](int buf, int k0) __attribute__((always_inline)) {
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int kk = 0; kk < BK / 2; ++kk)
        areg[buf][fm][kk] =
            A[im0 + wi0 + fm * 32 + r + (size_t)(k0 + kk * 2 + sub) * M];
  };
  auto load_b = [&](int k0) __attribute__((always_inline)) {
#pragma unroll
    for (int u = 0; u < BCH; ++u) {
      const int f = (u * 256 + tid) * 4;
      const int k = f / BN, j = f % BN;
      breg[u] = *(const f32x4*)(B + (jn0 + j) + (size_t)(k0 + k) * N);
    }
  };
  auto write_b = [&](int q) __attribute__((always_inline)) {
#pragma unroll
    for (int u = 0; u < BCH; ++u)
      *(f32x4*)(&Bs[q * BN * BK] + (u * 256 + tid) * 4) = breg[u];
  };

  const int niter = K / BK;
  // prologue: panel 0 into Bs[0] + areg[0]; panel 1 into breg
  load_b(0);
  write_b(0);
  load_a(0, 0);
  if (1 < niter) load_b(BK);
  __syncthreads();
  for (int it = 0; it < niter; ++it) {
    const int q = it & 1;
    // overlapped with this panel's MFMAs: publish panel it+1, fetch it+2
    if (it + 1 < niter) {
      write_b(q ^ 1);
      load_a((it & 1) ^ 1, (it + 1) * BK);
      if (it + 2 < niter) load_b((it + 2) * BK);
    }
    const float* Bp = &Bs[q * BN * BK];
    const int ab = it & 1;
#pragma unroll
    for (int kk = 0; kk < BK / 2; ++kk) {
      const int kloc = kk * 2 + sub;
      float b[4];
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) b[fn] = Bp[kloc * BN + fn * 32 + r];
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              areg[ab][fm][kk], b[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();  // readers of Bs[q] done AND writes to Bs[q^1] visible
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

// reference kernel: naive tiled (reuse shipped library shape via simple loop)
__global__ void k_ref(int n, const float* A, const float* B, float* C) {
  // one thread per C element, k-loop (slow; only for correctness at small n)
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  int j = blockIdx.y;
  if (i >= n) return;
  float s = 0.f;
  for (int k = 0; k < n; ++k) s += A[i + (size_t)k * n] * B[j + (size_t)k * n];
  C[i + (size_t)j * n] = s;
}

#define HIP_CALL(x)                                                   \
  do {                                                                \
    hipError_t e_ = (x);                                              \
    if (e_ != hipSuccess) {                                           \
      printf("HIP err %s @%d\n", hipGetErrorString(e_), __LINE__);    \
      return 1;                                                       \
    }                                                                 \
  } while (0)

template <int BK, int V>  // V: 0/1 = k_dtva single/double LDS, 2 = k_dtva2
static void run(const char* name, int n, const float* dA, const float* dB,
                float* dC, const float* dRef, float* dMax, int reps) {
  dim3 grid(n / BM, n / BN), block(256);
  auto launch1 = [&](float al, float be) {
    if (V == 2)
      hipLaunchKernelGGL((k_dtva2<BK>), grid, block, 0, 0, n, n, n, dA, dB,
                         dC, al, be);
    else
      hipLaunchKernelGGL((k_dtva<BK, V == 1>), grid, block, 0, 0, n, n, n,
                         dA, dB, dC, al, be);
  };
  hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
  launch1(1.f, 0.f);
  float md = 0.f;
  hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
  hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256), dim3(256),
                     0, 0, dC, dRef, (size_t)n * n, dMax);
  hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
  hipEvent_t b0, b1;
  hipEventCreate(&b0);
  hipEventCreate(&b1);
  for (int w = 0; w < 2; ++w) launch1(1.f, -1.5f);
  hipDeviceSynchronize();
  hipEventRecord(b0);
  for (int rr = 0; rr < reps; ++rr) launch1(1.f, -1.5f);
  hipEventRecord(b1);
  hipEventSynchronize(b1);
  float ms;
  hipEventElapsedTime(&ms, b0, b1);
  double gf = 2.0 * n * n * n * reps / (ms * 1e-3) / 1e9;
  printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n, name, gf, md,
         hipGetErrorString(hipGetLastError()));
  fflush(stdout);
}

int main(int argc, char** argv) {
  setbuf(stdout, NULL);
  int n = argc > 1 ? atoi(argv[1]) : 4096;
  const int reps = 10;
  size_t nn = (size_t)n * n, bytes = nn * 4;
  float *dA, *dB, *dC, *dRef, *dMax;
  HIP_CALL(hipMalloc(&dA, bytes));
  HIP_CALL(hipMalloc(&dB, bytes));
  HIP_CALL(hipMalloc(&dC, bytes));
  HIP_CALL(hipMalloc(&dRef, bytes));
  HIP_CALL(hipMalloc(&dMax, 4));
  hipLaunchKernelGGL(fill_lcg, dim3((nn + 255) / 256), dim3(256), 0, 0, dA,
                     nn, 1u);
  hipLaunchKernelGGL(fill_lcg, dim3((nn + 255) / 256), dim3(256), 0, 0, dB,
                     nn, 2u);
  hipLaunchKernelGGL(k_ref, dim3((n + 255) / 256, n), dim3(256), 0, 0, n, dA,
                     dB, dRef);
  HIP_CALL(hipDeviceSynchronize());
  for (int round = 0; round < 3; ++round) {
    printf("--- round %d\n", round);
    run<16, 0>("D1 dtva 256x128x16 1LDSB", n, dA, dB, dC, dRef, dMax, reps);
    run<16, 2>("D6 dtva2 256x128x16 1-barrier", n, dA, dB, dC, dRef, dMax,
               reps);
    run<8, 2>("D7 dtva2 256x128x8  1-barrier", n, dA, dB, dC, dRef, dMax,
              reps);
    run<32, 2>("D8 dtva2 256x128x32 1-barrier", n, dA, dB, dC, dRef, dMax,
               reps);
  }
  return 0;
}
