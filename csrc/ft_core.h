// ft_core.h — shared host API between the torch extension and the ft_sgemm
// CLI binary.
#pragma once

#include <hip/hip_runtime.h>

namespace ftsgemm {

// Launch one of the six hand-tiled MFMA SGEMM kernels (tier id per
// generated/tile_params.h; abft selects the fused-ABFT twin, inject the
// always-on fault injector).  C = alpha*A*B^T + beta*C, column-major.
// When abft, `ws` must point to a device scratch buffer of
// sgemm_abft_workspace_floats(tier, M, N, K) fp32 elements (the launcher
// fills it with the segment checksums of A and B before the fused kernel).
hipError_t sgemm_tier_launch(int tier, bool abft, bool inject, int M, int N,
                             int K, const float* A, const float* B, float* C,
                             float alpha, float beta, float tau,
                             float inj_mag, int verify_windows, float* ws,
                             hipStream_t stream);

bool sgemm_tier_supported(int tier, int M, int N, int K);

size_t sgemm_abft_workspace_floats(int tier, int M, int N, int K);

// rocBLAS paths (kernel id 0 oracle and the id-10 non-fused ABFT baseline,
// reference: cuBLAS at sgemm.cu:108 / include/baseline_ft_sgemm.cuh).
// Implemented in rocblas_path.hip.
struct BaselineWorkspace {
  // device buffers, all fp32; caller owns allocation
  float* ones;     // max(M, N) ones
  float* row_c;    // M   (C row sums)
  float* col_c;    // N   (C column sums)
  float* s_a;      // panel_k (A-panel column sums)
  float* s_b;      // panel_k (B-panel row sums)
  float* ref_row;  // M   (maintained row checksum)
  float* ref_col;  // N   (maintained col checksum)
  float* d_res;    // 2 * ceil(K / panel_k)  (device-side dot verdict slot
                   // pair per verified panel, stream-ordered; the host
                   // reduces to the WORST panel's verdicts)
};

int rocblas_sgemm_nt(int M, int N, int K, const float* A, const float* B,
                     float* C, float alpha, float beta, hipStream_t stream);

// Non-fused ABFT baseline: per 256-wide K panel, rocBLAS
// sgemm + 6x sgemv + 2x saxpy + 2x sdot (call-chain parity with
// baseline_ft_sgemm.cuh:3-32).  Returns 0 on success; verdicts (squared
// residual norms) written to res_row/res_col (host pointers).
int baseline_ft_sgemm(int M, int N, int K, const float* A, const float* B,
                      float* C, float alpha, float beta,
                      const BaselineWorkspace& ws, int panel_k,
                      float* res_row, float* res_col, hipStream_t stream);

}  // namespace ftsgemm
