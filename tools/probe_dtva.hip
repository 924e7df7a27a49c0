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



// k_dtvasm: the Tensile-structure SGEMM with ALL VMEM in inline asm so
// hipcc's conservative per-element vmcnt tracking (36 near-zero waits per
// 64 MFMAs in k_dtva2, measured) is replaced by two hand-counted waits per
// panel.  BM=256 BN=128, 4 waves of 64x128 (FM=2,FN=4); A slices are
// wave-private and go straight to registers; only B transits LDS.
// Guide form (ii): "=&v" loads in one statement, then a wait statement
// naming every destination "+v" before the first consumer (s_nop 1 for the
// VALU->MFMA boundary hazard).  BKT in {8, 16}.
template <int BKT>
__global__ __launch_bounds__(256, 2) void k_dtvasm(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  __shared__ __attribute__((aligned(16))) float Bs[2 * 128 * BKT];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * 256, jn0 = blockIdx.y * 128;

  constexpr int NA = BKT;          // A dwords per panel per lane (2fm*BKT/2)
  constexpr int NB = BKT / 8;      // B dwordx4 per panel per thread
  constexpr int NV = NA + NB;      // VMEM ops per panel

  f32x16 acc[2][4] = {};
  float aA[NA], aB[NA];
  f32x4 breg[NB];

  const int fB = tid * 4;          // B chunk 0; chunk 1 (BKT=16) at +1024
  const float* pB0 = B + (jn0 + (fB & 127)) + (size_t)(fB >> 7) * N;
  const float* pA0 = A + (im0 + wi0 + r) + (size_t)sub * M;

  // A fragment (fm,kk) register index: ao[fm*4+kk] per 8-k half; the
  // offset:128 forms load fm=1 (row +32 floats)

#define ISSUE8H(pb, q0, q1, q2, q3, b0v, aoff)                             \
  asm volatile("global_load_dwordx4 %0, %9, off\n\t"                        \
               "global_load_dword %1, %10, off\n\t"                         \
               "global_load_dword %2, %11, off\n\t"                         \
               "global_load_dword %3, %12, off\n\t"                         \
               "global_load_dword %4, %13, off\n\t"                         \
               "global_load_dword %5, %10, off offset:128\n\t"              \
               "global_load_dword %6, %11, off offset:128\n\t"              \
               "global_load_dword %7, %12, off offset:128\n\t"              \
               "global_load_dword %8, %13, off offset:128"                  \
               : "=&v"(b0v), "=&v"((aoff)[0]), "=&v"((aoff)[1]),            \
                 "=&v"((aoff)[2]), "=&v"((aoff)[3]), "=&v"((aoff)[4]),      \
                 "=&v"((aoff)[5]), "=&v"((aoff)[6]), "=&v"((aoff)[7])       \
               : "v"(pb), "v"(q0), "v"(q1), "v"(q2), "v"(q3))

#define ISSUE(p, bo, ao)                                                    \
  do {                                                                      \
    const size_t ko = (size_t)(p) * BKT;                                    \
    const float* pb = pB0 + ko * N;                                         \
    const float* q0 = pA0 + ko * M;                                         \
    const float* q1 = q0 + 2 * (size_t)M;                                   \
    const float* q2 = q0 + 4 * (size_t)M;                                   \
    const float* q3 = q0 + 6 * (size_t)M;                                   \
    ISSUE8H(pb, q0, q1, q2, q3, (bo)[0], ao);                               \
    if constexpr (BKT == 16) {                                              \
      const float* pb2 = pb + 8 * (size_t)N;                                \
      const float* q4 = q0 + 8 * (size_t)M;                                 \
      const float* q5 = q0 + 10 * (size_t)M;                                \
      const float* q6 = q0 + 12 * (size_t)M;                                \
      const float* q7 = q0 + 14 * (size_t)M;                                \
      ISSUE8H(pb2, q4, q5, q6, q7, (bo)[NB - 1], (ao) + 8);                 \
    }                                                                       \
  } while (0)

#define WAIT_A(n, ao)                                                       \
  do {                                                                      \
    asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                       \
                 : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),             \
                   "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),             \
                   "+v"((ao)[6]), "+v"((ao)[7]));                           \
    if constexpr (BKT == 16)                                                \
      asm volatile("" : "+v"((ao)[8]), "+v"((ao)[9]), "+v"((ao)[10]),       \
                     "+v"((ao)[11]), "+v"((ao)[12]), "+v"((ao)[13]),        \
                     "+v"((ao)[14]), "+v"((ao)[15]));                       \
  } while (0)
#define WAIT_B(n, bo)                                                       \
  do {                                                                      \
    asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1" : "+v"((bo)[0]));     \
    if constexpr (BKT == 16) asm volatile("" : "+v"((bo)[1]));              \
  } while (0)

  const int niter = K / BKT;
  ISSUE(0, breg, aA);
  WAIT_A(0, aA);  // prologue: drain everything (breg implied)
  WAIT_B(0, breg);
  *(f32x4*)(&Bs[0] + fB) = breg[0];
  if constexpr (BKT == 16) *(f32x4*)(&Bs[0] + fB + 1024) = breg[1];
  __syncthreads();

#define KKLOOP(q, ao)                                                       \
  do {                                                                      \
    const float* Bp = &Bs[(q) * 128 * BKT];                                 \
    _Pragma("unroll") for (int kk = 0; kk < BKT / 2; ++kk) {                \
      const int kloc = kk * 2 + sub;                                        \
      float b[4];                                                           \
      _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                      \
          b[fn] = Bp[kloc * 128 + fn * 32 + r];                             \
      _Pragma("unroll") for (int fm = 0; fm < 2; ++fm)                      \
          _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                  \
              acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(           \
                  (ao)[(kk / 4) * 8 + fm * 4 + (kk % 4)], b[fn],            \
                  acc[fm][fn], 0, 0, 0);                                    \
    }                                                                       \
  } while (0)

#define BODY(it, acur, anxt)                                                \
  do {                                                                      \
    const int q = (it) & 1;                                                 \
    if ((it) + 1 < niter) {                                                 \
      ISSUE((it) + 1, breg, anxt);                                          \
      if constexpr (BKT == 8) WAIT_A(9, acur);                              \
      else WAIT_A(18, acur);                                                \
    } else {                                                                \
      WAIT_A(0, acur);                                                      \
    }                                                                       \
    KKLOOP(q, acur);                                                        \
    if ((it) + 1 < niter) {                                                 \
      if constexpr (BKT == 8) {                                             \
        WAIT_B(8, breg);                                                    \
        *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = breg[0];                 \
      } else {                                                              \
        WAIT_B(8, breg); /* retires B0(pos1) and B1(pos10) of 18 */         \
        *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = breg[0];                 \
        *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB + 1024) = breg[1];          \
      }                                                                     \
    }                                                                       \
    __syncthreads();                                                        \
  } while (0)

  for (int it = 0; it < niter; it += 2) {
    BODY(it, aA, aB);
    BODY(it + 1, aB, aA);
  }
#undef ISSUE8
#undef ISSUE
#undef WAIT_A
#undef WAIT_B
#undef KKLOOP
#undef BODY

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


// k_dtvasm2: depth-2 VMEM pipeline — panels it+1 and it+2 stay in flight
// across the barrier; ONE hand-counted vmcnt(8) per panel covers both the
// B publish and the A consume (waits are "<=N outstanding", so the count
// is entry-state independent).  A register sets rotate with period 3,
// B staging regs with period 2 -> 6-body unroll with guarded tails.
__global__ __launch_bounds__(256, 1) void k_dtvasm2(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  constexpr int BKT = 8;
  __shared__ __attribute__((aligned(16))) float Bs[2 * 128 * BKT];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int sub = lane >> 5, r = lane & 31;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * 256, jn0 = blockIdx.y * 128;

  f32x16 acc[2][4] = {};
  float aA[8], aB[8], aC[8];
  f32x4 br0, br1;  // panel p's staged B quarter lives in br[p&1]

  const int fB = tid * 4;
  // SRSRC buffer addressing (T8): one descriptor per operand in SGPRs,
  // per-lane 32-bit voffset, per-panel k-advance through the SGPR soffset.
  const auto rsA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)A, (short)0, 0xffffffffu, 0x00020000);
  const auto rsB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)B, (short)0, 0xffffffffu, 0x00020000);
  const unsigned voffA = 4u * (unsigned)(im0 + wi0 + r + sub * (size_t)M);
  const unsigned voffB =
      4u * (unsigned)(jn0 + (fB & 127) + (size_t)(fB >> 7) * N);
  const unsigned sM8 = 8u * (unsigned)M;   // 2*M floats in bytes
  const unsigned sN32 = 32u * (unsigned)N; // BKT*N floats in bytes

#define ISSUE(p, b0v, ao)                                                   \
  do {                                                                      \
    const unsigned kB = (unsigned)(p) * sN32;                               \
    const unsigned k0 = (unsigned)(p) * BKT * 4u * (unsigned)M;             \
    const unsigned k1 = k0 + sM8;                                           \
    const unsigned k2 = k0 + 2 * sM8;                                       \
    const unsigned k3 = k0 + 3 * sM8;                                       \
    asm volatile("buffer_load_dwordx4 %0, %9, %10, %11 offen\n\t"          \
                 "buffer_load_dword %1, %12, %13, %14 offen\n\t"           \
                 "buffer_load_dword %2, %12, %13, %15 offen\n\t"           \
                 "buffer_load_dword %3, %12, %13, %16 offen\n\t"           \
                 "buffer_load_dword %4, %12, %13, %17 offen\n\t"           \
                 "buffer_load_dword %5, %12, %13, %14 offen offset:128\n\t" \
                 "buffer_load_dword %6, %12, %13, %15 offen offset:128\n\t" \
                 "buffer_load_dword %7, %12, %13, %16 offen offset:128\n\t" \
                 "buffer_load_dword %8, %12, %13, %17 offen offset:128"     \
                 : "=&v"(b0v), "=&v"((ao)[0]), "=&v"((ao)[1]),              \
                   "=&v"((ao)[2]), "=&v"((ao)[3]), "=&v"((ao)[4]),          \
                   "=&v"((ao)[5]), "=&v"((ao)[6]), "=&v"((ao)[7])           \
                 : "v"(voffB), "s"(rsB), "s"(kB), "v"(voffA), "s"(rsA),     \
                   "s"(k0), "s"(k1), "s"(k2), "s"(k3));                     \
  } while (0)
#define WAIT_AB(n, ao, bv)                                                  \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                         \
               : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),               \
                 "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),               \
                 "+v"((ao)[6]), "+v"((ao)[7]), "+v"(bv))
#define WAIT_A(n, ao)                                                       \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                         \
               : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),               \
                 "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),               \
                 "+v"((ao)[6]), "+v"((ao)[7]))
#define WAIT_B(n, bv) \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1" : "+v"(bv))

#define KKLOOP(q, ao)                                                       \
  do {                                                                      \
    const float* Bp = &Bs[(q) * 128 * BKT];                                 \
    _Pragma("unroll") for (int kk = 0; kk < BKT / 2; ++kk) {                \
      const int kloc = kk * 2 + sub;                                        \
      float b[4];                                                           \
      _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                      \
          b[fn] = Bp[kloc * 128 + fn * 32 + r];                             \
      _Pragma("unroll") for (int fm = 0; fm < 2; ++fm)                      \
          _Pragma("unroll") for (int fn = 0; fn < 4; ++fn)                  \
              acc[fm][fn] = __builtin_amdgcn_mfma_f32_32x32x2f32(           \
                  (ao)[fm * 4 + kk], b[fn], acc[fm][fn], 0, 0, 0);          \
    }                                                                       \
  } while (0)

  // body it: consume Acur + Bs[it&1]; publish BRwr (panel it+1's B) to
  // Bs[(it+1)&1]; issue panel it+2 into (BRiss, Anxt).
#define BODY(it, Acur, Anxt, BRwr, BRiss)                                   \
  do {                                                                      \
    const int q = (it) & 1;                                                 \
    if ((it) + 2 < niter) {                                                 \
      WAIT_AB(8, Acur, BRwr);                                               \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRwr;                      \
      ISSUE((it) + 2, BRiss, Anxt);                                         \
    } else if ((it) + 1 < niter) {                                          \
      WAIT_AB(8, Acur, BRwr);                                               \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRwr;                      \
    } else {                                                                \
      WAIT_A(0, Acur);                                                      \
    }                                                                       \
    KKLOOP(q, Acur);                                                        \
    __syncthreads();                                                        \
  } while (0)

  const int niter = K / BKT;  // caller guarantees niter >= 3
  ISSUE(0, br0, aA);
  ISSUE(1, br1, aB);
  WAIT_B(17, br0);
  *(f32x4*)(&Bs[0] + fB) = br0;
  __syncthreads();

  int it = 0;
  while (it < niter) {
    BODY(it, aA, aC, br1, br0);
    if (++it >= niter) break;
    BODY(it, aB, aA, br0, br1);
    if (++it >= niter) break;
    BODY(it, aC, aB, br1, br0);
    if (++it >= niter) break;
    BODY(it, aA, aC, br0, br1);
    if (++it >= niter) break;
    BODY(it, aB, aA, br1, br0);
    if (++it >= niter) break;
    BODY(it, aC, aB, br0, br1);
    ++it;
  }
#undef ISSUE
#undef WAIT_AB
#undef WAIT_A
#undef WAIT_B
#undef KKLOOP
#undef BODY

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



// k_dtvasm16: the D9 asm skeleton with rocBLAS's MFMA choice
// (v_mfma_f32_16x16x4_f32): fragment rows land on 16-lane groups, so the
// four fm offsets become immediate `offset:` fields (3 pointer inputs).
// PAD (D15): pad each B k-row by PAD*16 floats.  Unpadded, the row stride
// (128 floats) is 0 mod 64 banks, so the four 16-lane s16 groups of a
// wave hit the SAME 16 banks on every ds_read_b32 of B — a 4-way
// conflict 8x per k-step.  PAD=1 (stride 144 ≡ 16 mod 64) lands each s16
// group on its own 16-bank quarter: conflict-free reads.  This is
// exactly Tensile's LPB16 (LDS pad B 16) in the rocBLAS kernel name —
// and it is only possible with register->ds_write staging: the glds
// (HBM->LDS DMA) path of the shipped kernel writes LDS contiguously and
// cannot pad.
template <int OCC = 2, int PAD = 0>
__global__ __launch_bounds__(256, OCC) void k_dtvasm16(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  constexpr int BKT = 8;
  constexpr int LS = 128 + PAD * 16;  // padded LDS row stride (floats)
  __shared__ __attribute__((aligned(16))) float Bs[2 * LS * BKT];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int s16 = lane >> 4, r16 = lane & 15;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * 256, jn0 = blockIdx.y * 128;

  f32x4 acc[4][8] = {};
  float aA[8], aB[8];
  f32x4 breg;

  const int fB = tid * 4;
  const float* pB0 = B + (jn0 + (fB & 127)) + (size_t)(fB >> 7) * N;
  const float* pA0 = A + (im0 + wi0 + r16) + (size_t)s16 * M;

#define ISSUE(p, b0v, ao)                                                   \
  do {                                                                      \
    const size_t ko = (size_t)(p) * BKT;                                    \
    const float* pb = pB0 + ko * N;                                         \
    const float* q0 = pA0 + ko * M;                                         \
    const float* q1 = q0 + 4 * (size_t)M;                                   \
    asm volatile("global_load_dwordx4 %0, %9, off\n\t"                      \
                 "global_load_dword %1, %10, off\n\t"                       \
                 "global_load_dword %2, %10, off offset:64\n\t"             \
                 "global_load_dword %3, %10, off offset:128\n\t"            \
                 "global_load_dword %4, %10, off offset:192\n\t"            \
                 "global_load_dword %5, %11, off\n\t"                       \
                 "global_load_dword %6, %11, off offset:64\n\t"             \
                 "global_load_dword %7, %11, off offset:128\n\t"            \
                 "global_load_dword %8, %11, off offset:192"                \
                 : "=&v"(b0v), "=&v"((ao)[0]), "=&v"((ao)[1]),              \
                   "=&v"((ao)[2]), "=&v"((ao)[3]), "=&v"((ao)[4]),          \
                   "=&v"((ao)[5]), "=&v"((ao)[6]), "=&v"((ao)[7])           \
                 : "v"(pb), "v"(q0), "v"(q1));                              \
  } while (0)
#define WAIT_A(n, ao)                                                       \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                         \
               : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),               \
                 "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),               \
                 "+v"((ao)[6]), "+v"((ao)[7]))
#define WAIT_B(n) \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1" : "+v"(breg))

  const int niter = K / BKT;
  ISSUE(0, breg, aA);
  WAIT_B(8);
  WAIT_A(0, aA);
  *(f32x4*)(&Bs[0] + (fB >> 7) * LS + (fB & 127)) = breg;
  __syncthreads();

#define KKLOOP(q, ao)                                                       \
  do {                                                                      \
    const float* Bp = &Bs[(q) * LS * BKT];                                 \
    _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) {                      \
      const int kloc = kk * 4 + s16;                                        \
      float b[8];                                                           \
      _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                      \
          b[fn] = Bp[kloc * LS + fn * 16 + r16];                           \
      _Pragma("unroll") for (int fm = 0; fm < 4; ++fm)                      \
          _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                  \
              acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x4f32(           \
                  (ao)[kk * 4 + fm], b[fn], acc[fm][fn], 0, 0, 0);          \
    }                                                                       \
  } while (0)

#define BODY(it, acur, anxt)                                                \
  do {                                                                      \
    const int q = (it) & 1;                                                 \
    if ((it) + 1 < niter) {                                                 \
      ISSUE((it) + 1, breg, anxt);                                          \
      WAIT_A(9, acur);                                                      \
    } else {                                                                \
      WAIT_A(0, acur);                                                      \
    }                                                                       \
    KKLOOP(q, acur);                                                        \
    if ((it) + 1 < niter) {                                                 \
      WAIT_B(8);                                                            \
      *(f32x4*)(&Bs[(q ^ 1) * LS * BKT] + (fB >> 7) * LS + (fB & 127)) = breg;                      \
    }                                                                       \
    __syncthreads();                                                        \
  } while (0)

  for (int it = 0; it < niter; it += 2) {
    BODY(it, aA, aB);
    BODY(it + 1, aB, aA);
  }
#undef ISSUE
#undef WAIT_A
#undef WAIT_B
#undef KKLOOP
#undef BODY

#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 8; ++fn) {
      const int j = jn0 + fn * 16 + r16;
      float* p = C + (size_t)j * M + im0 + wi0 + fm * 16 + s16 * 4;
      const f32x4 prev = *(const f32x4*)p;
      f32x4 out;
#pragma unroll
      for (int u = 0; u < 4; ++u)
        out[u] = alpha * acc[fm][fn][u] + beta * prev[u];
      *(f32x4*)p = out;
    }
}


// k_dtvasm16d2 (D14): the D12 skeleton with a DEPTH-2 VMEM pipeline —
// panels it+1 AND it+2 stay in flight across the barrier, so the per-panel
// A-wait (vmcnt(18)) and B-wait (vmcnt(17)) are satisfied ~always and the
// only hard sync left is the barrier + lgkm drain.  Register budget:
// D12's 232 + one more A set (8) + one more B staging reg (4) = 244 < 256
// -> still 2 waves/SIMD (the 512-entry unified file / 8-granule rule,
// MI355X_MICROARCH.md §register file).  A sets rotate with period 3, B
// staging with period 2 -> 6-body unrolled loop (requires niter % 6 == 0
// handled by the guarded while, niter >= 2).
template <int OCC = 2>
__global__ __launch_bounds__(256, OCC) void k_dtvasm16d2(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  constexpr int BKT = 8;
  __shared__ __attribute__((aligned(16))) float Bs[2 * 128 * BKT];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int s16 = lane >> 4, r16 = lane & 15;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * 256, jn0 = blockIdx.y * 128;

  f32x4 acc[4][8] = {};
  float aA[8], aB[8], aC[8];
  f32x4 br0, br1;

  const int fB = tid * 4;
  const float* pB0 = B + (jn0 + (fB & 127)) + (size_t)(fB >> 7) * N;
  const float* pA0 = A + (im0 + wi0 + r16) + (size_t)s16 * M;

#define ISSUE(p, b0v, ao)                                                   \
  do {                                                                      \
    const size_t ko = (size_t)(p) * BKT;                                    \
    const float* pb = pB0 + ko * N;                                         \
    const float* q0 = pA0 + ko * M;                                         \
    const float* q1 = q0 + 4 * (size_t)M;                                   \
    asm volatile("global_load_dwordx4 %0, %9, off\n\t"                      \
                 "global_load_dword %1, %10, off\n\t"                       \
                 "global_load_dword %2, %10, off offset:64\n\t"             \
                 "global_load_dword %3, %10, off offset:128\n\t"            \
                 "global_load_dword %4, %10, off offset:192\n\t"            \
                 "global_load_dword %5, %11, off\n\t"                       \
                 "global_load_dword %6, %11, off offset:64\n\t"             \
                 "global_load_dword %7, %11, off offset:128\n\t"            \
                 "global_load_dword %8, %11, off offset:192"                \
                 : "=&v"(b0v), "=&v"((ao)[0]), "=&v"((ao)[1]),              \
                   "=&v"((ao)[2]), "=&v"((ao)[3]), "=&v"((ao)[4]),          \
                   "=&v"((ao)[5]), "=&v"((ao)[6]), "=&v"((ao)[7])           \
                 : "v"(pb), "v"(q0), "v"(q1));                              \
  } while (0)
#define WAIT_A(n, ao)                                                       \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                         \
               : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),               \
                 "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),               \
                 "+v"((ao)[6]), "+v"((ao)[7]))
#define WAIT_B(n, bv) \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1" : "+v"(bv))

  const int niter = K / BKT;
  ISSUE(0, br0, aA);
  ISSUE(1, br1, aB);
  WAIT_B(17, br0);  // B0 retired (A0 x8 + batch1 x9 issued after it)
  *(f32x4*)(&Bs[0] + fB) = br0;
  __syncthreads();

#define KKLOOP(q, ao)                                                       \
  do {                                                                      \
    const float* Bp = &Bs[(q) * 128 * BKT];                                 \
    _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) {                      \
      const int kloc = kk * 4 + s16;                                        \
      float b[8];                                                           \
      _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                      \
          b[fn] = Bp[kloc * 128 + fn * 16 + r16];                           \
      _Pragma("unroll") for (int fm = 0; fm < 4; ++fm)                      \
          _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                  \
              acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x4f32(           \
                  (ao)[kk * 4 + fm], b[fn], acc[fm][fn], 0, 0, 0);          \
    }                                                                       \
  } while (0)

  // body it: Bs[q] holds B_it (published last body); Acur = A_it (in
  // flight from 2 bodies ago); BRwr = B_{it+1} staging reg; issue panel
  // it+2 into (BRiss, Anxt).  Publish B_{it+1} BEFORE the MFMA loop so the
  // lgkm drain overlaps the MFMA stream instead of stalling the barrier.
#define BODY(it, Acur, Anxt, BRwr, BRiss)                                   \
  do {                                                                      \
    const int q = (it) & 1;                                                 \
    if ((it) + 2 < niter) {                                                 \
      ISSUE((it) + 2, BRiss, Anxt);                                         \
      WAIT_B(17, BRwr);                                                     \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRwr;                      \
      WAIT_A(18, Acur);                                                     \
    } else if ((it) + 1 < niter) {                                          \
      WAIT_B(8, BRwr);                                                      \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRwr;                      \
      WAIT_A(9, Acur);                                                      \
    } else {                                                                \
      WAIT_A(0, Acur);                                                      \
    }                                                                       \
    KKLOOP(q, Acur);                                                        \
    __syncthreads();                                                        \
  } while (0)

  int it = 0;
  while (it < niter) {
    BODY(it, aA, aC, br1, br0);
    if (++it >= niter) break;
    BODY(it, aB, aA, br0, br1);
    if (++it >= niter) break;
    BODY(it, aC, aB, br1, br0);
    if (++it >= niter) break;
    BODY(it, aA, aC, br0, br1);
    if (++it >= niter) break;
    BODY(it, aB, aA, br1, br0);
    if (++it >= niter) break;
    BODY(it, aC, aB, br0, br1);
    ++it;
  }
#undef ISSUE
#undef WAIT_A
#undef WAIT_B
#undef KKLOOP
#undef BODY

#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 8; ++fn) {
      const int j = jn0 + fn * 16 + r16;
      float* p = C + (size_t)j * M + im0 + wi0 + fm * 16 + s16 * 4;
      const f32x4 prev = *(const f32x4*)p;
      f32x4 out;
#pragma unroll
      for (int u = 0; u < 4; ++u)
        out[u] = alpha * acc[fm][fn][u] + beta * prev[u];
      *(f32x4*)p = out;
    }
}


// k_dtvasm16ag (D16): the D12 skeleton with a DEPTH-2 VMEM pipeline —
// panels it+1 AND it+2 stay in flight across the barrier, so the per-panel
// A-wait (vmcnt(18)) and B-wait (vmcnt(17)) are satisfied ~always and the
// only hard sync left is the barrier + lgkm drain.  Register budget:
// D12's 232 + one more A set (8) + one more B staging reg (4) = 244 < 256
// -> still 2 waves/SIMD (the 512-entry unified file / 8-granule rule,
// MI355X_MICROARCH.md §register file).  A sets rotate with period 3, B
// staging with period 2 -> 6-body unrolled loop (requires niter % 6 == 0
// handled by the guarded while, niter >= 2).
template <int OCC = 2>
__global__ __launch_bounds__(256, OCC) void k_dtvasm16ag(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  constexpr int BKT = 8;
  __shared__ __attribute__((aligned(16))) float Bs[2 * 128 * BKT];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int s16 = lane >> 4, r16 = lane & 15;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * 256, jn0 = blockIdx.y * 128;

  f32x4 acc[4][8] = {};
  float aA[8], aB[8], aC[8];
  f32x4 br0, br1;

  const int fB = tid * 4;
  const float* pB0 = B + (jn0 + (fB & 127)) + (size_t)(fB >> 7) * N;
  const float* pA0 = A + (im0 + wi0 + r16) + (size_t)s16 * M;

#define ISSUE(p, b0v, ao)                                                   \
  do {                                                                      \
    const size_t ko = (size_t)(p) * BKT;                                    \
    const float* pb = pB0 + ko * N;                                         \
    const float* q0 = pA0 + ko * M;                                         \
    const float* q1 = q0 + 4 * (size_t)M;                                   \
    asm volatile("global_load_dwordx4 %0, %9, off\n\t"                      \
                 "global_load_dword %1, %10, off\n\t"                       \
                 "global_load_dword %2, %10, off offset:64\n\t"             \
                 "global_load_dword %3, %10, off offset:128\n\t"            \
                 "global_load_dword %4, %10, off offset:192\n\t"            \
                 "global_load_dword %5, %11, off\n\t"                       \
                 "global_load_dword %6, %11, off offset:64\n\t"             \
                 "global_load_dword %7, %11, off offset:128\n\t"            \
                 "global_load_dword %8, %11, off offset:192"                \
                 : "=&v"(b0v), "=&v"((ao)[0]), "=&v"((ao)[1]),              \
                   "=&v"((ao)[2]), "=&v"((ao)[3]), "=&v"((ao)[4]),          \
                   "=&v"((ao)[5]), "=&v"((ao)[6]), "=&v"((ao)[7])           \
                 : "v"(pb), "v"(q0), "v"(q1));                              \
  } while (0)
#define WAIT_A(n, ao)                                                       \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                         \
               : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),               \
                 "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),               \
                 "+v"((ao)[6]), "+v"((ao)[7]))
#define WAIT_B(n, bv) \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1" : "+v"(bv))

  const int niter = K / BKT;
  ISSUE(0, br0, aA);
  ISSUE(1, br1, aB);
  WAIT_B(17, br0);  // B0 retired (A0 x8 + batch1 x9 issued after it)
  *(f32x4*)(&Bs[0] + fB) = br0;
  __syncthreads();

#define KKLOOP(q, ao)                                                       \
  do {                                                                      \
    const float* Bp = &Bs[(q) * 128 * BKT];                                 \
    _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) {                      \
      const int kloc = kk * 4 + s16;                                        \
      float b[8];                                                           \
      _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                      \
          b[fn] = Bp[kloc * 128 + fn * 16 + r16];                           \
      _Pragma("unroll") for (int fm = 0; fm < 4; ++fm)                      \
          _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                  \
              asm volatile("v_mfma_f32_16x16x4_f32 %0, %1, %2, %0"          \
                           : "+a"(acc[fm][fn])                              \
                           : "v"((ao)[kk * 4 + fm]), "v"(b[fn]));           \
    }                                                                       \
  } while (0)

  // body it: Bs[q] holds B_it (published last body); Acur = A_it (in
  // flight from 2 bodies ago); BRwr = B_{it+1} staging reg; issue panel
  // it+2 into (BRiss, Anxt).  Publish B_{it+1} BEFORE the MFMA loop so the
  // lgkm drain overlaps the MFMA stream instead of stalling the barrier.
#define BODY(it, Acur, Anxt, BRwr, BRiss)                                   \
  do {                                                                      \
    const int q = (it) & 1;                                                 \
    if ((it) + 2 < niter) {                                                 \
      ISSUE((it) + 2, BRiss, Anxt);                                         \
      WAIT_B(17, BRwr);                                                     \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRwr;                      \
      WAIT_A(18, Acur);                                                     \
    } else if ((it) + 1 < niter) {                                          \
      WAIT_B(8, BRwr);                                                      \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRwr;                      \
      WAIT_A(9, Acur);                                                      \
    } else {                                                                \
      WAIT_A(0, Acur);                                                      \
    }                                                                       \
    KKLOOP(q, Acur);                                                        \
    __syncthreads();                                                        \
  } while (0)

  int it = 0;
  while (it < niter) {
    BODY(it, aA, aC, br1, br0);
    if (++it >= niter) break;
    BODY(it, aB, aA, br0, br1);
    if (++it >= niter) break;
    BODY(it, aC, aB, br1, br0);
    if (++it >= niter) break;
    BODY(it, aA, aC, br0, br1);
    if (++it >= niter) break;
    BODY(it, aB, aA, br1, br0);
    if (++it >= niter) break;
    BODY(it, aC, aB, br0, br1);
    ++it;
  }
#undef ISSUE
#undef WAIT_A
#undef WAIT_B
#undef KKLOOP
#undef BODY

  asm volatile("s_nop 7" ::: "memory");  // MFMA D -> VALU-reader hazard
#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 8; ++fn) {
      const int j = jn0 + fn * 16 + r16;
      float* p = C + (size_t)j * M + im0 + wi0 + fm * 16 + s16 * 4;
      const f32x4 prev = *(const f32x4*)p;
      f32x4 out;
#pragma unroll
      for (int u = 0; u < 4; ++u)
        out[u] = alpha * acc[fm][fn][u] + beta * prev[u];
      *(f32x4*)p = out;
    }
}



// k_dtvasm16p2 (D17b): depth-2 at D12's register budget via SPLIT issues,
// all period-2 (the first D17 cut reused the published B register for the
// next issue, which collapsed the rotation to one register and produced
// wrong panels; D14/D16's 3-set rotation spilled 2.4-2.8 KB/lane).
//   top(it):  ISSUE_B(it+2) -> br[it&1]   (B_it published last body ->
//             register free; B wait slack = 1 full body)
//   wait vmcnt(9): retires B_{it+1} (and everything older, incl. A_it
//             which has 2 bodies of slack); publish B_{it+1} to LDS
//   KKLOOP(A_it); ISSUE_A(it+2) -> a[it&1] (just consumed); barrier.
template <int OCC = 2, bool SAFE = false>
__global__ __launch_bounds__(256, OCC) void k_dtvasm16p2(
    int M, int N, int K, const float* __restrict__ A,
    const float* __restrict__ B, float* __restrict__ C, float alpha,
    float beta) {
  constexpr int BKT = 8;
  __shared__ __attribute__((aligned(16))) float Bs[2 * 128 * BKT];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int s16 = lane >> 4, r16 = lane & 15;
  const int wi0 = wave * 64;
  const int im0 = blockIdx.x * 256, jn0 = blockIdx.y * 128;

  f32x4 acc[4][8] = {};
  float aA[8], aB[8];
  f32x4 br0, br1;

  const int fB = tid * 4;
  const float* pB0 = B + (jn0 + (fB & 127)) + (size_t)(fB >> 7) * N;
  const float* pA0 = A + (im0 + wi0 + r16) + (size_t)s16 * M;

#define ISSUE(p, b0v, ao)                                                   \
  do {                                                                      \
    const size_t ko = (size_t)(p) * BKT;                                    \
    const float* pb = pB0 + ko * N;                                         \
    const float* q0 = pA0 + ko * M;                                         \
    const float* q1 = q0 + 4 * (size_t)M;                                   \
    asm volatile("global_load_dwordx4 %0, %9, off\n\t"                      \
                 "global_load_dword %1, %10, off\n\t"                       \
                 "global_load_dword %2, %10, off offset:64\n\t"             \
                 "global_load_dword %3, %10, off offset:128\n\t"            \
                 "global_load_dword %4, %10, off offset:192\n\t"            \
                 "global_load_dword %5, %11, off\n\t"                       \
                 "global_load_dword %6, %11, off offset:64\n\t"             \
                 "global_load_dword %7, %11, off offset:128\n\t"            \
                 "global_load_dword %8, %11, off offset:192"                \
                 : "=&v"(b0v), "=&v"((ao)[0]), "=&v"((ao)[1]),              \
                   "=&v"((ao)[2]), "=&v"((ao)[3]), "=&v"((ao)[4]),          \
                   "=&v"((ao)[5]), "=&v"((ao)[6]), "=&v"((ao)[7])           \
                 : "v"(pb), "v"(q0), "v"(q1));                              \
  } while (0)
#define WAIT_A(n, ao)                                                       \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1"                         \
               : "+v"((ao)[0]), "+v"((ao)[1]), "+v"((ao)[2]),               \
                 "+v"((ao)[3]), "+v"((ao)[4]), "+v"((ao)[5]),               \
                 "+v"((ao)[6]), "+v"((ao)[7]))
#define WAIT_B(n, bv) \
  asm volatile("s_waitcnt vmcnt(" #n ")\n\ts_nop 1" : "+v"(bv))

#define ISSUE_B(p, b0v)                                                     \
  do {                                                                      \
    const float* pb = pB0 + (size_t)(p) * BKT * N;                          \
    asm volatile("global_load_dwordx4 %0, %1, off"                          \
                 : "=&v"(b0v) : "v"(pb));                                   \
  } while (0)
#define ISSUE_A(p, ao)                                                      \
  do {                                                                      \
    const size_t ko = (size_t)(p) * BKT;                                    \
    const float* q0 = pA0 + ko * M;                                         \
    const float* q1 = q0 + 4 * (size_t)M;                                   \
    asm volatile("global_load_dword %0, %8, off\n\t"                        \
                 "global_load_dword %1, %8, off offset:64\n\t"              \
                 "global_load_dword %2, %8, off offset:128\n\t"             \
                 "global_load_dword %3, %8, off offset:192\n\t"             \
                 "global_load_dword %4, %9, off\n\t"                        \
                 "global_load_dword %5, %9, off offset:64\n\t"              \
                 "global_load_dword %6, %9, off offset:128\n\t"             \
                 "global_load_dword %7, %9, off offset:192"                 \
                 : "=&v"((ao)[0]), "=&v"((ao)[1]), "=&v"((ao)[2]),          \
                   "=&v"((ao)[3]), "=&v"((ao)[4]), "=&v"((ao)[5]),          \
                   "=&v"((ao)[6]), "=&v"((ao)[7])                           \
                 : "v"(q0), "v"(q1));                                       \
  } while (0)

  const int niter = K / BKT;
  ISSUE_B(0, br0);
  ISSUE_A(0, aA);
  if (1 < niter) {
    ISSUE_B(1, br1);
    ISSUE_A(1, aB);
  }
  WAIT_B(17, br0);  // B0 retired (A0 x8 + B1 + A1 x8 = 17 after it)
  *(f32x4*)(&Bs[0] + fB) = br0;
  __syncthreads();

#define KKLOOP(q, ao)                                                       \
  do {                                                                      \
    const float* Bp = &Bs[(q) * 128 * BKT];                                 \
    _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) {                      \
      const int kloc = kk * 4 + s16;                                        \
      float b[8];                                                           \
      _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                      \
          b[fn] = Bp[kloc * 128 + fn * 16 + r16];                           \
      _Pragma("unroll") for (int fm = 0; fm < 4; ++fm)                      \
          _Pragma("unroll") for (int fn = 0; fn < 8; ++fn)                  \
              acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x4f32(           \
                  (ao)[kk * 4 + fm], b[fn], acc[fm][fn], 0, 0, 0);          \
    }                                                                       \
  } while (0)

  // body it: Bs[it&1] holds B_it (published last body); Acur = A_it (in
  // flight since body it-2 -> arrived); BRwr = B_{it+1} (issued last
  // body, 1-body slack).  Publish B_{it+1} BEFORE the MFMA loop; issue
  // panel it+2 AFTER it, into the registers this body just freed.
#define BODY(it, Acur, BRcur, BRnxt)                                        \
  do {                                                                      \
    const int q = (it) & 1;                                                 \
    if ((it) + 2 < niter) {                                                 \
      ISSUE_B((it) + 2, BRcur);                                             \
      if constexpr (SAFE) WAIT_A(0, Acur);                                  \
      else WAIT_A(10, Acur);                                                \
    } else if ((it) + 1 < niter) {                                          \
      WAIT_A(9, Acur);                                                      \
    } else {                                                                \
      WAIT_A(0, Acur);                                                      \
    }                                                                       \
    KKLOOP(q, Acur);                                                        \
    if ((it) + 1 < niter) {                                                 \
      if constexpr (SAFE) WAIT_B(0, BRnxt);                                 \
      else WAIT_B(2, BRnxt);                                                \
      *(f32x4*)(&Bs[(q ^ 1) * 128 * BKT] + fB) = BRnxt;                     \
    }                                                                       \
    if ((it) + 2 < niter) ISSUE_A((it) + 2, Acur);                          \
    __syncthreads();                                                        \
  } while (0)

  for (int it = 0; it < niter; it += 2) {
    BODY(it, aA, br0, br1);
    if (it + 1 < niter) BODY(it + 1, aB, br1, br0);
  }
#undef ISSUE_B
#undef ISSUE_A
#undef WAIT_A
#undef WAIT_B
#undef KKLOOP
#undef BODY

#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int fn = 0; fn < 8; ++fn) {
      const int j = jn0 + fn * 16 + r16;
      float* p = C + (size_t)j * M + im0 + wi0 + fm * 16 + s16 * 4;
      const f32x4 prev = *(const f32x4*)p;
      f32x4 out;
#pragma unroll
      for (int u = 0; u < 4; ++u)
        out[u] = alpha * acc[fm][fn][u] + beta * prev[u];
      *(f32x4*)p = out;
    }
}

__global__ void fill_lcg(float* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  unsigned s = seed ^ (unsigned)(i * 2654435761u);
  s ^= s << 13; s ^= s >> 17; s ^= s << 5;
  p[i] = ((s >> 8) * (1.0f / 16777216.0f)) * 1.8f - 0.9f;
}

// bad-element census for debugging: counts |diff|>0.1 and records the
// first few linear indices (column-major C: idx = i + j*n)
__global__ void bad_census(const float* x, const float* y, size_t n,
                           unsigned* count, unsigned* idxs) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  if (fabsf(x[i] - y[i]) > 0.1f) {
    unsigned slot = atomicAdd(count, 1u);
    if (slot < 16) idxs[slot] = (unsigned)i;
  }
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
    {  // asm-pipelined variant
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm<8>, grid, block, 0, 0, n, n, n, dA, dB, dC,
                         1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm<8>, grid, block, 0, 0, n, n, n, dA, dB, dC,
                           1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm<8>, grid, block, 0, 0, n, n, n, dA, dB, dC,
                           1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D9 dtvasm 256x128x8 asm-vmem",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // depth-2 asm pipeline
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm2, grid, block, 0, 0, n, n, n, dA, dB, dC,
                         1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm2, grid, block, 0, 0, n, n, n, dA, dB,
                           dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm2, grid, block, 0, 0, n, n, n, dA, dB,
                           dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D10 dtvasm2 depth-2 pipeline",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // BK=16 asm pipeline
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm<16>, grid, block, 0, 0, n, n, n, dA, dB,
                         dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm<16>, grid, block, 0, 0, n, n, n, dA, dB,
                           dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm<16>, grid, block, 0, 0, n, n, n, dA, dB,
                           dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D11 dtvasm 256x128x16 asm-vmem",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // 16x16x4 MFMA on the asm skeleton
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm16<2>, grid, block, 0, 0, n, n, n, dA, dB, dC,
                         1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm16<2>, grid, block, 0, 0, n, n, n, dA, dB,
                           dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm16<2>, grid, block, 0, 0, n, n, n, dA, dB,
                           dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D12 dtvasm16 16x16x4 asm",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // D15: D12 + LDS-pad-B (Tensile LPB16): conflict-free B reads
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL((k_dtvasm16<2, 1>), grid, block, 0, 0, n, n, n, dA,
                         dB, dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL((k_dtvasm16<2, 1>), grid, block, 0, 0, n, n, n,
                           dA, dB, dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL((k_dtvasm16<2, 1>), grid, block, 0, 0, n, n, n,
                           dA, dB, dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D15 dtvasm16 LPB16 padded",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // depth-2 pipeline on the 16x16x4 asm skeleton (D14)
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm16d2<2>, grid, block, 0, 0, n, n, n, dA, dB,
                         dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm16d2<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm16d2<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D14 dtvasm16 depth-2",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // D17: period-2 depth-2 (issue at body end into freed regs)
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm16p2<2>, grid, block, 0, 0, n, n, n, dA,
                         dB, dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm16p2<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm16p2<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      {
        unsigned *dCnt;
        hipMalloc(&dCnt, 4 + 16 * 4);
        hipMemsetAsync(dCnt, 0, 4 + 16 * 4, 0);
        hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
        hipLaunchKernelGGL(k_dtvasm16p2<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, 0.f);
        hipLaunchKernelGGL(bad_census, dim3(((size_t)n * n + 255) / 256),
                           dim3(256), 0, 0, dC, dRef, (size_t)n * n, dCnt,
                           dCnt + 1);
        unsigned h[17];
        hipMemcpy(h, dCnt, sizeof h, hipMemcpyDeviceToHost);
        printf("D17 bad elements: %u of %zu; first idx (i,j): ", h[0],
               (size_t)n * n);
        for (int u = 0; u < 8 && u < (int)h[0]; ++u)
          printf("(%u,%u) ", h[1 + u] % n, h[1 + u] / n);
        printf("\n");
        hipFree(dCnt);
      }
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D17 dtvasm16 p2 depth-2",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // D17-SAFE: same rotation, full-drain waits (wait-count bisect)
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL((k_dtvasm16p2<2, true>), grid, block, 0, 0, n, n,
                         n, dA, dB, dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL((k_dtvasm16p2<2, true>), grid, block, 0, 0, n,
                           n, n, dA, dB, dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL((k_dtvasm16p2<2, true>), grid, block, 0, 0, n,
                           n, n, dA, dB, dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D17S dtvasm16 p2 SAFE-waits",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // D16: depth-2 + AGPR accumulator
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm16ag<2>, grid, block, 0, 0, n, n, n, dA,
                         dB, dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm16ag<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm16ag<2>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D16 AGPR depth-2 (spills, see ABLATION)",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
    {  // OCC=3 variant (3 blocks/CU if it fits 168 VGPRs)
      dim3 grid(n / BM, n / BN), block(256);
      hipMemsetD32Async((hipDeviceptr_t)dC, 0, (size_t)n * n, 0);
      hipLaunchKernelGGL(k_dtvasm16<3>, grid, block, 0, 0, n, n, n, dA, dB,
                         dC, 1.f, 0.f);
      float md = 0.f;
      hipMemcpyAsync(dMax, &md, 4, hipMemcpyHostToDevice, 0);
      hipLaunchKernelGGL(max_diff, dim3(((size_t)n * n + 255) / 256),
                         dim3(256), 0, 0, dC, dRef, (size_t)n * n, dMax);
      hipMemcpy(&md, dMax, 4, hipMemcpyDeviceToHost);
      hipEvent_t b0, b1;
      hipEventCreate(&b0);
      hipEventCreate(&b1);
      for (int w = 0; w < 2; ++w)
        hipLaunchKernelGGL(k_dtvasm16<3>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipDeviceSynchronize();
      hipEventRecord(b0);
      for (int rr = 0; rr < reps; ++rr)
        hipLaunchKernelGGL(k_dtvasm16<3>, grid, block, 0, 0, n, n, n, dA,
                           dB, dC, 1.f, -1.5f);
      hipEventRecord(b1);
      hipEventSynchronize(b1);
      float ms;
      hipEventElapsedTime(&ms, b0, b1);
      printf("N=%d %-30s %8.0f GFLOPS (maxdiff %.2e) err=%s\n", n,
             "D13 dtvasm16 OCC3",
             2.0 * n * n * n * reps / (ms * 1e-3) / 1e9, md,
             hipGetErrorString(hipGetLastError()));
    }
  }
  return 0;
}
