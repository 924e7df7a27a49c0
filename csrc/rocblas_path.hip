// rocblas_path.hip — rocBLAS-composed paths: kernel-0 oracle GEMM and the
// non-fused ABFT baseline (kernel id 10).
//
// Reference parity: cuBLAS oracle at sgemm.cu:108/260 and the 11-call
// cuBLAS chain of include/baseline_ft_sgemm.cuh:1-33 (per 256-wide K panel:
// 1 gemm + 6 gemv + 2 axpy + 2 dot).  Unlike the reference (whose per-panel
// checksum comparison is only meaningful for the first panel), the
// maintained checksums here accumulate across panels, so the verdict
// scalars are genuinely ~0 for a fault-free GEMM at any K, and the
// RETURNED verdict is the WORST panel's, not the last's (each panel's dot
// lands in its own d_res slot — ADVICE r01 #4).
//
// Two modes (FT_SGEMM_BASELINE_MODE):
//   "chain" — strict call-mapping parity: every checksum op is a rocBLAS
//             call (sgemv/saxpy/sdot), 1:1 with the reference chain.
//   default — same maths, but the four bandwidth-bound reductions go
//             through hand-written CDNA4 kernels (one-pass row/col sums,
//             f32x4, wave+LDS reduce) instead of rocblas_sgemv: the
//             per-panel full-C gemv sweeps dominated the measured 34.3%
//             overhead in round 1.
// FT_SGEMM_VERIFY_EVERY=j (default 1) runs the C-sweep + verdict every
// j-th panel (and always on the last): the maintained operand checksums
// still update EVERY panel, so a detected fault is located in time at a
// j-panel granularity — the knob trades verdict latency for overhead
// (VERDICT r01 allows per-j-panel verdicts; the reference verifies per
// panel at T4 cost ratios).

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <roctracer/roctx.h>

#include <cstdlib>
#include <cstring>
#include <mutex>
#include <vector>

#include "ft_core.h"

namespace ftsgemm {

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;

// out[j] = sum_i X[i + j*rows] for j in [0, cols): column sums of a
// column-major rows x cols matrix.  One workgroup per column, f32x4 down
// the contiguous column, butterfly + LDS reduce.  Serves both the C column
// sweep (cols = N) and the panel operand sums s_a/s_b (cols = panel_k).
__global__ __launch_bounds__(256) void colsum_kernel(
    int rows, const float* __restrict__ X, float* __restrict__ out) {
  const int j = blockIdx.x;
  const float* col = X + (size_t)j * rows;
  const int tid = threadIdx.x;
  float s = 0.f;
  const int vrows = rows & ~3;
  if ((rows & 3) == 0) {
    for (int i = tid * 4; i < vrows; i += 1024) {
      const f32x4 v = *(const f32x4*)(col + i);
      s += (v[0] + v[1]) + (v[2] + v[3]);
    }
  } else {  // odd fallback shapes (rocBLAS-fallback path): scalar
    for (int i = tid; i < rows; i += 256) s += col[i];
  }
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) s += __shfl_xor(s, m, 64);
  __shared__ float partial[4];
  if ((tid & 63) == 0) partial[tid >> 6] = s;
  __syncthreads();
  if (tid == 0)
    out[j] = (partial[0] + partial[1]) + (partial[2] + partial[3]);
}

// out[i] += sum_j X[i + j*rows] over this block's column slice: row sums
// of a column-major rows x cols matrix.  Grid = (row bands, column
// slices): enough workgroups to fill 256 CUs even for one M=4096 sweep
// (a single-axis rows/1024 grid was 4 workgroups — measured 10x slower
// than the rocblas gemv it replaced).  Each (band, slice) workgroup
// accumulates its 1024-row f32x4 partial over JCH columns and combines
// across slices with one unsafeAtomicAdd per element — rows*gridDim.y
// atomics total, ~1 MB at N=4096.  Caller zero-fills `out` first.
constexpr int ROWSUM_JCH = 64;  // columns per slice
__global__ __launch_bounds__(256) void rowsum_kernel(
    int rows, int cols, const float* __restrict__ X,
    float* __restrict__ out) {
  const int i = blockIdx.x * 1024 + threadIdx.x * 4;
  const int j0 = blockIdx.y * ROWSUM_JCH;
  const int j1 = (j0 + ROWSUM_JCH < cols) ? j0 + ROWSUM_JCH : cols;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  if ((rows & 3) == 0 && i + 4 <= rows) {
    for (int j = j0; j < j1; ++j) {
      const f32x4 v = *(const f32x4*)(X + (size_t)j * rows + i);
#pragma unroll
      for (int u = 0; u < 4; ++u) acc[u] += v[u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) unsafeAtomicAdd(out + i + u, acc[u]);
  } else {
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      if (i + u < rows) {
        float s = 0.f;
        for (int j = j0; j < j1; ++j) s += X[(size_t)j * rows + i + u];
        unsafeAtomicAdd(out + i + u, s);
      }
    }
  }
}

int env_int(const char* name, int dflt) {
  const char* e = std::getenv(name);
  return e ? std::atoi(e) : dflt;
}

bool chain_mode() {
  const char* e = std::getenv("FT_SGEMM_BASELINE_MODE");
  return e && std::strcmp(e, "chain") == 0;
}

}  // namespace

static rocblas_handle get_handle() {
  static rocblas_handle h = nullptr;
  static std::once_flag once;
  std::call_once(once, [] { rocblas_create_handle(&h); });
  return h;
}

#define RB_CHECK(x)                                \
  do {                                             \
    rocblas_status st_ = (x);                      \
    if (st_ != rocblas_status_success) return (int)st_; \
  } while (0)

int rocblas_sgemm_nt(int M, int N, int K, const float* A, const float* B,
                     float* C, float alpha, float beta, hipStream_t stream) {
  rocblas_handle h = get_handle();
  RB_CHECK(rocblas_set_stream(h, stream));
  RB_CHECK(rocblas_sgemm(h, rocblas_operation_none,
                         rocblas_operation_transpose, M, N, K, &alpha, A, M,
                         B, N, &beta, C, M));
  return 0;
}

int baseline_ft_sgemm(int M, int N, int K, const float* A, const float* B,
                      float* C, float alpha, float beta,
                      const BaselineWorkspace& ws, int panel_k,
                      float* res_row, float* res_col, hipStream_t stream) {
  rocblas_handle h = get_handle();
  RB_CHECK(rocblas_set_stream(h, stream));
  // Host pointer mode for the alpha/beta scalars; the per-panel dot
  // verdicts go to DEVICE memory (ws.d_res slots) so the chain stays fully
  // stream-ordered — a host-pointer sdot would synchronize every panel
  // (the reference pays that sync at baseline_ft_sgemm.cuh:28,31; on
  // MI355X it costs more than the dot itself).
  RB_CHECK(rocblas_set_pointer_mode(h, rocblas_pointer_mode_host));
  const float one = 1.f, zero = 0.f, neg1 = -1.f;
  const bool chain = chain_mode();
  // default j=2: the C-sweep verdict runs every 2nd panel (and always on
  // the last), halving the dominant full-C read cost; the maintained
  // operand checksums still update EVERY panel, so end-of-GEMM detection
  // coverage is unchanged — only intermediate verdict latency coarsens
  // (measured at 4096: j=1 33.5%, j=2 28.4% overhead vs same-run rocBLAS;
  // reference ratio 30.1%).  FT_SGEMM_VERIFY_EVERY=1 restores per-panel
  // verdicts.
  int vevery = env_int("FT_SGEMM_VERIFY_EVERY", 2);
  if (vevery < 1) vevery = 1;

  // Initialise maintained checksums with the beta*C contribution so the
  // verdict stays ~0 for any beta (the alpha factor is folded into the
  // per-panel gemv updates below).
  RB_CHECK(rocblas_sgemv(h, rocblas_operation_none, M, N, &beta, C, M,
                         ws.ones, 1, &zero, ws.ref_row, 1));
  RB_CHECK(rocblas_sgemv(h, rocblas_operation_transpose, M, N, &beta, C, M,
                         ws.ones, 1, &zero, ws.ref_col, 1));

  const int npanels = (K + panel_k - 1) / panel_k;
  int pv = 0;  // verdict slot counter (2 floats per verdict)
  for (int p = 0; p < npanels; ++p) {
    const int k0 = p * panel_k;
    const int kp = (K - k0 < panel_k) ? (K - k0) : panel_k;
    const float* Ap = A + (size_t)k0 * M;
    const float* Bp = B + (size_t)k0 * N;
    const float beta_run = (k0 == 0) ? beta : 1.f;
    // Panel GEMM: C = alpha * Ap * Bp^T + beta_run * C
    roctxRangePush("baseline_panel_gemm");
    RB_CHECK(rocblas_sgemm(h, rocblas_operation_none,
                           rocblas_operation_transpose, M, N, kp, &alpha, Ap,
                           M, Bp, N, &beta_run, C, M));
    roctxRangePop();
    roctxRangePush("baseline_checksum_update");
    // Panel operand sums: s_a = Ap^T e_M, s_b = Bp^T e_N
    if (chain) {
      RB_CHECK(rocblas_sgemv(h, rocblas_operation_transpose, M, kp, &one, Ap,
                             M, ws.ones, 1, &zero, ws.s_a, 1));
      RB_CHECK(rocblas_sgemv(h, rocblas_operation_transpose, N, kp, &one, Bp,
                             N, ws.ones, 1, &zero, ws.s_b, 1));
    } else {
      hipLaunchKernelGGL(colsum_kernel, dim3(kp), dim3(256), 0, stream, M,
                         Ap, ws.s_a);
      hipLaunchKernelGGL(colsum_kernel, dim3(kp), dim3(256), 0, stream, N,
                         Bp, ws.s_b);
    }
    // Maintained checksums: ref_row += alpha * Ap s_b ; ref_col += alpha * Bp s_a
    RB_CHECK(rocblas_sgemv(h, rocblas_operation_none, M, kp, &alpha, Ap, M,
                           ws.s_b, 1, &one, ws.ref_row, 1));
    RB_CHECK(rocblas_sgemv(h, rocblas_operation_none, N, kp, &alpha, Bp, N,
                           ws.s_a, 1, &one, ws.ref_col, 1));
    roctxRangePop();  // baseline_checksum_update
    if ((p + 1) % vevery != 0 && p != npanels - 1) continue;
    roctxRangePush("baseline_verify_verdict");
    // Observed sums of the running C
    if (chain) {
      RB_CHECK(rocblas_sgemv(h, rocblas_operation_none, M, N, &one, C, M,
                             ws.ones, 1, &zero, ws.row_c, 1));
      RB_CHECK(rocblas_sgemv(h, rocblas_operation_transpose, M, N, &one, C,
                             M, ws.ones, 1, &zero, ws.col_c, 1));
    } else {
      if (hipMemsetAsync(ws.row_c, 0, M * sizeof(float), stream) !=
          hipSuccess)
        return -1;
      hipLaunchKernelGGL(rowsum_kernel,
                         dim3((M + 1023) / 1024,
                              (N + ROWSUM_JCH - 1) / ROWSUM_JCH),
                         dim3(256), 0, stream, M, N, C, ws.row_c);
      hipLaunchKernelGGL(colsum_kernel, dim3(N), dim3(256), 0, stream, M, C,
                         ws.col_c);
    }
    // Residual + scalar verdict (axpy + dot, as the reference does at
    // baseline_ft_sgemm.cuh:25-31), verdicts stream-ordered into device
    // memory, one slot pair per verified panel
    RB_CHECK(rocblas_saxpy(h, M, &neg1, ws.ref_row, 1, ws.row_c, 1));
    RB_CHECK(rocblas_saxpy(h, N, &neg1, ws.ref_col, 1, ws.col_c, 1));
    RB_CHECK(rocblas_set_pointer_mode(h, rocblas_pointer_mode_device));
    RB_CHECK(rocblas_sdot(h, M, ws.row_c, 1, ws.row_c, 1, ws.d_res + 2 * pv));
    RB_CHECK(
        rocblas_sdot(h, N, ws.col_c, 1, ws.col_c, 1, ws.d_res + 2 * pv + 1));
    RB_CHECK(rocblas_set_pointer_mode(h, rocblas_pointer_mode_host));
    roctxRangePop();  // baseline_verify_verdict
    ++pv;
  }
  std::vector<float> res(2 * pv);
  if (hipMemcpyAsync(res.data(), ws.d_res, 2 * pv * sizeof(float),
                     hipMemcpyDeviceToHost, stream) != hipSuccess ||
      hipStreamSynchronize(stream) != hipSuccess)
    return -1;
  *res_row = 0.f;
  *res_col = 0.f;
  for (int v = 0; v < pv; ++v) {  // worst panel, not last
    if (res[2 * v] > *res_row) *res_row = res[2 * v];
    if (res[2 * v + 1] > *res_col) *res_col = res[2 * v + 1];
  }
  return 0;
}

}  // namespace ftsgemm
