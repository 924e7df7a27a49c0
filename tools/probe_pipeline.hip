// probe_pipeline.hip — pipeline-structure ablation for the PLAIN MFMA SGEMM
// (the 126.7 TF vs ~147 TF achievable gap at 4096^3 fp32).
//
// Variants (within-probe A/B, random operands, self-verified vs variant 0):
//   P0  BK=32 2-buf, __syncthreads at panel end        (shipped structure)
//   P1  BK=32 2-buf, raw s_barrier at panel TOP + counted vmcnt, stage after
//       the barrier ("write tile t+1 after the barrier" form)
//   P2  BK=64 2-buf, __syncthreads                     (1 block/CU, 128 KB LDS)
//   P3  BK=32 3-buf ring, raw barrier + vmcnt(8) span  (1 block/CU, 96 KB LDS)
//   P4  P1 + XCD-aware bijective blockIdx swizzle
//   P5  P2 + raw barrier form
// Sizes: 4096 (L3-fit) and 8192 (past L3) to separate issue-bound from
// HBM/L2-bound effects.
//
// Build: hipcc -x hip --offload-arch=gfx950 -O3 tools/probe_pipeline.hip -o bin/probe_pipeline

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../csrc/ft_kernels.hpp"  // library kernel, for wave-shape variants

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;

#define BM 128
#define BN 128
#define WM 64
#define WN 64

__device__ inline void raw_barrier() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
}

// STRUCT: 0 = syncthreads-at-end (shipped), 1 = raw barrier at top + stage
// after barrier, 2 = 3-buffer ring with one panel left in flight.
template <int BK, int STRUCT, bool SWIZ, bool PRIO = false>
__global__ __launch_bounds__(256) void k_pipe(int M, int N, int K,
                                              const float* __restrict__ A,
                                              const float* __restrict__ B,
                                              float* __restrict__ C,
                                              float alpha, float beta) {
  constexpr int BUF = (BM + BN) * BK;
  constexpr int NBUF = (STRUCT == 2) ? 3 : 2;
  __shared__ __attribute__((aligned(16))) float lds[NBUF * BUF];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wm_idx = wave >> 1, wn_idx = wave & 1;
  const int wi0 = wm_idx * WM, wj0 = wn_idx * WN;

  int bx = blockIdx.x, by = blockIdx.y;
  if constexpr (SWIZ) {
    // bijective XCD remap over the flattened grid (guide §5 template):
    // consecutive new ids land on one XCD.
    const int nwg = gridDim.x * gridDim.y;
    const int orig = by * gridDim.x + bx;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = orig % 8;
    const int wgid = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) +
                     orig / 8;
    bx = wgid % gridDim.x;
    by = wgid / gridDim.x;
  }
  const int im0 = bx * BM, jn0 = by * BN;

  f32x16 acc[2][2] = {};

  constexpr int GA = (BM * BK) / (256 * 4);
  constexpr int GB = (BN * BK) / (256 * 4);
  auto stage = [&](int q, int k0) __attribute__((always_inline)) {
#pragma unroll
    for (int t = 0; t < GA; ++t) {
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
    for (int t = 0; t < GB; ++t) {
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
  constexpr int LD = GA + GB;  // glds instructions per panel

  auto kloop = [&](int q) __attribute__((always_inline)) {
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
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              a[fm], b[fn], acc[fm][fn], 0, 0, 0);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
    }
  };

  const int niter = K / BK;
  if constexpr (STRUCT == 0) {
    stage(0, 0);
    __syncthreads();
    for (int it = 0; it < niter; ++it) {
      const int q = it & 1;
      if (it + 1 < niter) stage(q ^ 1, (it + 1) * BK);
      kloop(q);
      __syncthreads();
    }
  } else if constexpr (STRUCT == 3) {
    // Register-pipelined: fragments for kk+1 load while kk computes, and the
    // FIRST fragment of the next panel loads BEFORE the barrier (its buffer
    // already landed — vmcnt-drained), so every wave has MFMA-ready
    // operands the moment the barrier releases.  Raw s_barrier, no waits:
    // all reads of the outgoing buffer were consumed by MFMAs (lgkm waits
    // already emitted); the outstanding prefetch targets the incoming
    // buffer, which nobody overwrites for another two panels.
    auto frag_load = [&](const float* As, const float* Bs, int kk, float* af,
                         float* bf) __attribute__((always_inline)) {
      const int kloc = kk * 2 + sub;
      af[0] = As[kloc * BM + wi0 + r];
      af[1] = As[kloc * BM + wi0 + 32 + r];
      bf[0] = Bs[kloc * BN + wj0 + r];
      bf[1] = Bs[kloc * BN + wj0 + 32 + r];
    };
    stage(0, 0);
    __syncthreads();
    float af[2][2], bf[2][2];
    frag_load(&lds[0], &lds[0] + BM * BK, 0, af[0], bf[0]);
    for (int it = 0; it < niter; ++it) {
      const int q = it & 1;
      const float* As = &lds[q * BUF];
      const float* Bs = As + BM * BK;
      const float* An = &lds[(q ^ 1) * BUF];
      const float* Bn = An + BM * BK;
      if (it + 1 < niter) stage(q ^ 1, (it + 1) * BK);
#pragma unroll
      for (int kk = 0; kk < BK / 2; ++kk) {
        const int cur = kk & 1, nxt = cur ^ 1;
        if (kk + 1 < BK / 2) {
          frag_load(As, Bs, kk + 1, af[nxt], bf[nxt]);
        } else if (it + 1 < niter) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
          frag_load(An, Bn, 0, af[nxt], bf[nxt]);
        }
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                af[cur][fm], bf[cur][fn], acc[fm][fn], 0, 0, 0);
      }
      __builtin_amdgcn_s_barrier();
    }
  } else if constexpr (STRUCT == 1) {
    // 2-buffer span: barrier first (frees q^1), issue next stage, THEN a
    // counted wait that drains only panel it — panel it+1 stays in flight
    // across the whole kloop.
    stage(0, 0);
    for (int it = 0; it < niter; ++it) {
      const int q = it & 1;
      raw_barrier();  // readers of q^1 (panel it-1) are done everywhere
      if (it + 1 < niter) {
        stage(q ^ 1, (it + 1) * BK);
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(LD) : "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      kloop(q);
    }
  } else {  // STRUCT == 2: 3-buffer ring, panels it+1 and it+2 in flight
    stage(0, 0);
    stage(1, BK);
    for (int it = 0; it < niter; ++it) {
      const int q = it % 3;
      raw_barrier();  // readers of buffer (it+2)%3 (panel it-1) done
      if (it + 2 < niter) {
        stage((it + 2) % 3, (it + 2) * BK);
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * LD) : "memory");
      } else if (it + 1 < niter) {
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(LD) : "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      kloop(q);
    }
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

__global__ void fill_lcg(float* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  unsigned s = seed ^ (unsigned)(i * 2654435761u);
  s ^= s << 13; s ^= s >> 17; s ^= s << 5;
  p[i] = ((s >> 8) * (1.0f / 16777216.0f)) * 1.8f - 0.9f;  // [-0.9, 0.9)
}

__global__ void max_diff(const float* x, const float* y, size_t n,
                         float* out) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  float d = 0.f;
  if (i < n) d = fabsf(x[i] - y[i]);
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) d = fmaxf(d, __shfl_xor(d, m, 64));
  if ((threadIdx.x & 63) == 0 && d > *out) atomicMax((int*)out, __float_as_int(d));
}

#define HIP_CALL(x)                                                 \
  do {                                                              \
    hipError_t e_ = (x);                                            \
    if (e_ != hipSuccess) {                                         \
      printf("HIP err %s @%d\n", hipGetErrorString(e_), __LINE__);  \
      return 1;                                                     \
    }                                                               \
  } while (0)

// Library-kernel tile/wave-shape variants (plain or fused-ABFT+inject).
template <int BM_, int BN_, int BK_, int WM_, int WN_, bool ABFT_ = false,
          bool INJ_ = false, bool NTC_ = false, bool SWIZ_ = false,
          int BETA10 = 0, int MM_ = 32, int OCC_ = 2, int PIPE_ = 0>
static void run_lib(const char* name, int n, const float* dA, const float* dB,
                    float* dC, const float* dRef, float* dMax, int reps) {
  const float beta_ = BETA10 / 10.f;
  dim3 grid(n / BM_, n / BN_), block(64 * (BM_ / WM_) * (BN_ / WN_));
  static float* ws = nullptr;
  const int sstr = (n + 63) & ~63;
  if (ABFT_ && !ws) hipMalloc(&ws, 2 * (size_t)(8192 / 64) * 8192 * 4);
  const int niter = n / BK_, stride = niter / 20 > 0 ? niter / 20 : 1;
  auto launch = [&]() {
    if (ABFT_) {
      hipLaunchKernelGGL((ftsgemm::segsum_kernel<WM_>), dim3(n), dim3(256),
                         0, 0, n, n, sstr, dA, ws,
                         ws + (size_t)(n / WM_) * sstr);
      hipLaunchKernelGGL(
          (ftsgemm::sgemm_mfma<BM_, BN_, BK_, WM_, WN_, MM_, ABFT_, INJ_,
                               NTC_, SWIZ_, OCC_>),
          grid, block, 0, 0, n, n, n, dA, dB, dC, 1.f, beta_, stride, stride,
          9500.f, 10000.f, ws, sstr);
    } else {
      hipLaunchKernelGGL(
          (ftsgemm::sgemm_mfma<BM_, BN_, BK_, WM_, WN_, MM_, false, false,
                               NTC_, SWIZ_, OCC_, PIPE_>),
          grid, block, 0, 0, n, n, n, dA, dB, dC, 1.f, beta_, stride, stride,
          1e30f, 0.f, nullptr, 0);
    }
  };
  hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
  launch();
  float md = 0.f;
  if (BETA10 == 0) {
    hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
    hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256), dim3(256),
                       0, 0, dC, dRef, (size_t)n * n, dMax);
    hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
  }
  hipEvent_t b0, b1;
  hipEventCreate(&b0);
  hipEventCreate(&b1);
  for (int w = 0; w < 2; ++w) launch();
  hipDeviceSynchronize();
  hipEventRecord(b0);
  for (int r = 0; r < reps; ++r) launch();
  hipEventRecord(b1);
  hipEventSynchronize(b1);
  float ms;
  hipEventElapsedTime(&ms, b0, b1);
  double gf = 2.0 * n * n * n * reps / (ms * 1e-3) / 1e9;
  printf("N=%d %-34s %8.0f GFLOPS  (maxdiff %.2e) err=%s\n", n, name, gf, md,
         hipGetErrorString(hipGetLastError()));
  fflush(stdout);
}

template <int BK, int STRUCT, bool SWIZ, bool PRIO = false>
static void run_variant(const char* name, int n, const float* dA,
                        const float* dB, float* dC, const float* dRef,
                        float* dMax, int reps) {
  dim3 grid(n / BM, n / BN), block(256);
  hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
  hipLaunchKernelGGL((k_pipe<BK, STRUCT, SWIZ, PRIO>), grid, block, 0, 0, n,
                     n, n, dA, dB, dC, 1.f, 0.f);
  float md = 0.f;
  if (dRef) {
    hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
    hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256), dim3(256),
                       0, 0, dC, dRef, (size_t)n * n, dMax);
    hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
  }
  hipEvent_t b0, b1;
  hipEventCreate(&b0);
  hipEventCreate(&b1);
  for (int w = 0; w < 2; ++w)
    hipLaunchKernelGGL((k_pipe<BK, STRUCT, SWIZ, PRIO>), grid, block, 0, 0, n,
                       n, n, dA, dB, dC, 1.f, 0.f);
  hipDeviceSynchronize();
  hipEventRecord(b0);
  for (int r = 0; r < reps; ++r)
    hipLaunchKernelGGL((k_pipe<BK, STRUCT, SWIZ, PRIO>), grid, block, 0, 0, n,
                       n, n, dA, dB, dC, 1.f, 0.f);
  hipEventRecord(b1);
  hipEventSynchronize(b1);
  float ms;
  hipEventElapsedTime(&ms, b0, b1);
  double gf = 2.0 * n * n * n * reps / (ms * 1e-3) / 1e9;
  printf("N=%d %-34s %8.0f GFLOPS  (maxdiff %.2e) err=%s\n", n, name, gf, md,
         hipGetErrorString(hipGetLastError()));
  fflush(stdout);
}

int main(int argc, char** argv) {
  setbuf(stdout, NULL);
  const int reps = 10;
  std::vector<int> sizes = {4096, 8192};
  if (argc > 1) { sizes = {atoi(argv[1])}; }
  for (int n : sizes) {
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
    // reference result from P0
    hipMemsetD32Async((hipDeviceptr_t)dRef, 0, nn, 0);
    {
      dim3 grid(n / BM, n / BN), block(256);
      hipLaunchKernelGGL((k_pipe<32, 0, false>), grid, block, 0, 0, n, n, n,
                         dA, dB, dRef, 1.f, 0.f);
    }
    HIP_CALL(hipDeviceSynchronize());
    const char* only = getenv("PROBE_ONLY");
    for (int round = 0; round < 2; ++round) {
      printf("--- N=%d round %d\n", n, round);
      if (only && *only == 'L') {
        run_lib<128, 128, 32, 64, 64>("PA lib shipped", n, dA, dB, dC, dRef,
                                      dMax, reps);
        continue;
      }
      if (only && *only == 'T') {
        // tier tuning: tall / wide / large variants (beta=-1.5)
        run_lib<128, 32, 32, 64, 32, false, false, false, false, -15>(
            "tall    bk32 plain (shipped)", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 32, 16, 64, 32, false, false, false, false, -15>(
            "tall    bk16 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 32, 32, 64, 32, true, true, false, false, -15>(
            "tall    bk32 abft+inj (shipped)", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<128, 32, 16, 64, 32, true, true, false, false, -15>(
            "tall    bk16 abft+inj", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 32, 16, 128, 32, true, true, false, false, -15>(
            "tall    bk16 w1 WM128 abft+inj", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<32, 128, 32, 32, 64, false, false, false, false, -15>(
            "wide    bk32 plain (shipped)", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 128, 16, 32, 64, false, false, false, false, -15>(
            "wide    bk16 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 128, 16, 32, 64, true, true, false, false, -15>(
            "wide    bk16 abft+inj", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 128, 16, 32, 128, true, true, false, false, -15>(
            "wide    bk16 w1 WN128 abft+inj", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<64, 64, 32, 64, 64, false, false, false, false, -15>(
            "large   bk32 plain (shipped)", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 64, 16, 64, 64, false, false, false, false, -15>(
            "large   bk16 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 64, 16, 64, 64, false, false, false, false, -15, 32, 4>(
            "large   bk16 occ4 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 64, 16, 64, 64, true, true, false, false, -15>(
            "large   bk16 abft+inj", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 64, 32, 64, 64, true, true, false, false, -15>(
            "large   bk32 abft+inj (shipped)", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<32, 32, 16, 32, 32, false, false, false, false, -15>(
            "medium  bk16 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 32, 16, 32, 32, true, true, false, false, -15>(
            "medium  bk16 abft+inj", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 32, 16, 64, 32, true, true, false, false, -15, 32, 3>(
            "tall    bk16 abft+inj OCC3", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 32, 16, 64, 32, true, true, false, false, -15, 32, 4>(
            "tall    bk16 abft+inj OCC4", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 128, 16, 32, 64, true, true, false, false, -15, 32, 3>(
            "wide    bk16 abft+inj OCC3", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 128, 16, 32, 64, true, true, false, false, -15, 32, 4>(
            "wide    bk16 abft+inj OCC4", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 64, 16, 64, 64, true, true, false, false, -15, 32, 3>(
            "large   bk16 abft+inj OCC3", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<32, 32, 16, 32, 32, true, true, false, false, -15, 32, 4>(
            "medium  bk16 abft+inj OCC4", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 64, 16, 32, 64, false, false, false, false, -15>(
            "large   bk16 w2 WM32xWN64 plain", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<64, 64, 16, 32, 64, true, true, false, false, -15>(
            "large   bk16 w2 WM32xWN64 abft+inj", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<256, 128, 16, 128, 64, true, true, false, false, -15>(
            "huge    bk16 abft+inj (enc-asm)", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<256, 128, 16, 128, 64, false, false, false, false, -15>(
            "huge    bk16 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 64, 16, 64, 64, false, false, false, false, -15>(
            "tall2   128x64x16 w2 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<128, 64, 16, 64, 64, true, true, false, false, -15>(
            "tall2   128x64x16 w2 abft+inj", n, dA, dB, dC, dRef, dMax,
            reps);
        run_lib<64, 128, 16, 64, 64, false, false, false, false, -15>(
            "wide2   64x128x16 w2 plain", n, dA, dB, dC, dRef, dMax, reps);
        run_lib<64, 128, 16, 64, 64, true, true, false, false, -15>(
            "wide2   64x128x16 w2 abft+inj", n, dA, dB, dC, dRef, dMax,
            reps);
        continue;
      }
      run_variant<32, 0, false>("P0 bk32 2buf syncthreads", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<32, 3, false>("P10 bk32 regpipe x-barrier", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<16, 3, false>("P11 bk16 regpipe x-barrier", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<64, 3, false>("P12 bk64 regpipe x-barrier", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<16, 0, false>("PB bk16 2buf syncthreads", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<16, 1, false>("PC bk16 2buf rawbar/top-wait", n, dA, dB,
                                dC, dRef, dMax, reps);
      run_variant<32, 1, false>("P1 bk32 2buf rawbar/top-wait", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<64, 0, false>("P2 bk64 2buf syncthreads", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<32, 2, false>("P3 bk32 3buf vmcnt-span", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<32, 1, true>("P4 = P1 + XCD swizzle", n, dA, dB, dC, dRef,
                               dMax, reps);
      run_variant<64, 1, false>("P5 bk64 2buf rawbar/top-wait", n, dA, dB, dC,
                                dRef, dMax, reps);
      run_variant<64, 1, true>("P6 = P5 + XCD swizzle", n, dA, dB, dC, dRef,
                               dMax, reps);
      run_variant<32, 1, false, true>("P7 = P1 + setprio(1) on MFMA", n, dA,
                                      dB, dC, dRef, dMax, reps);
      run_lib<128, 128, 32, 64, 128>("P8 lib 128x128x32 w2 WM64xWN128", n,
                                     dA, dB, dC, dRef, dMax, reps);
      run_lib<128, 128, 32, 128, 64>("P9 lib 128x128x32 w2 WM128xWN64", n,
                                     dA, dB, dC, dRef, dMax, reps);
      run_lib<128, 128, 32, 64, 64>("PA lib 128x128x32 w4 (shipped)", n, dA,
                                    dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 32, 64, 128>("T1 lib 256x128x32 w4 WM64xWN128", n,
                                     dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 64, 128>("T2 lib 256x128x16 w4 WM64xWN128", n,
                                     dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64>("T3 lib 256x128x16 w4 WM128xWN64", n,
                                     dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 8, 128, 64>("T6 lib 256x128x8 w4 WM128xWN64", n, dA,
                                    dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 32, 64, 64>("T7 lib 256x128x32 w8 WM64xWN64", n, dA,
                                    dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 64, 64>("T8 lib 256x128x16 w8 WM64xWN64", n, dA,
                                    dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, true, true>(
          "A1 abft+inj 256x128x16 w4 WM128xWN64", n, dA, dB, dC, dRef, dMax,
          reps);
      run_lib<256, 128, 16, 64, 128, true, true>(
          "A2 abft+inj 256x128x16 w4 WM64xWN128", n, dA, dB, dC, dRef, dMax,
          reps);
      run_lib<256, 128, 16, 64, 64, true, true>(
          "A3 abft+inj 256x128x16 w8 WM64xWN64", n, dA, dB, dC, dRef, dMax,
          reps);
      run_lib<128, 128, 16, 64, 64, true, true>(
          "A4 abft+inj 128x128x16 w4 WM64xWN64", n, dA, dB, dC, dRef, dMax,
          reps);
      run_lib<128, 128, 32, 64, 64, true, true>(
          "A5 abft+inj 128x128x32 w4 WM64xWN64", n, dA, dB, dC, dRef, dMax,
          reps);
      run_lib<256, 128, 16, 64, 64>("A3p plain  256x128x16 w8 WM64xWN64", n,
                                    dA, dB, dC, dRef, dMax, reps);
      // beta = -1.5 family (C read-modify-write traffic): NTC / SWIZ
      run_lib<256, 128, 16, 128, 64, false, false, false, false, -15>(
          "B0 plain b-1.5 base", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, true, false, -15>(
          "B1 plain b-1.5 +NTC", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, false, true, -15>(
          "B2 plain b-1.5 +SWIZ", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, true, true, -15>(
          "B3 plain b-1.5 +NTC+SWIZ", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, true, true, true, true, -15>(
          "B4 abft+inj b-1.5 +NTC+SWIZ", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, true, true, false, false, -15>(
          "B5 abft+inj b-1.5 base", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, true, true, 0>(
          "B6 plain b0 +NTC+SWIZ", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, false, false, -15, 16>(
          "C1 plain b-1.5 MM16 (16x16x4)", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, false, false, -15, 32, 3>(
          "C2 plain b-1.5 bk16 OCC3", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 8, 128, 64, false, false, false, false, -15, 32, 3>(
          "C3 plain b-1.5 bk8 OCC3", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 8, 128, 64, false, false, false, false, -15, 32, 2>(
          "C4 plain b-1.5 bk8 OCC2", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<128, 128, 16, 128, 64, false, false, false, false, -15, 32, 4>(
          "C5 plain b-1.5 128x128x16 w2 OCC4", n, dA, dB, dC, dRef, dMax,
          reps);
      run_lib<256, 128, 16, 128, 64, true, true, false, false, -15, 32, 3>(
          "C6 abft+inj b-1.5 bk16 OCC3", n, dA, dB, dC, dRef, dMax, reps);
      run_lib<256, 128, 16, 128, 64, false, false, false, false, -15, 32, 2,
              1>("E1 plain b-1.5 bk16 3buf-ring", n, dA, dB, dC, dRef, dMax,
                 reps);
      run_lib<256, 128, 8, 128, 64, false, false, false, false, -15, 32, 2,
              1>("E2 plain b-1.5 bk8 3buf-ring", n, dA, dB, dC, dRef, dMax,
                 reps);
      run_lib<256, 128, 32, 64, 128, false, false, false, false, -15, 32, 2,
              1>("E3 plain b-1.5 bk32 3buf-ring w4 WM64WN128", n, dA, dB,
                 dC, dRef, dMax, reps);
    }
    hipFree(dA); hipFree(dB); hipFree(dC); hipFree(dRef); hipFree(dMax);
  }
  return 0;
}
