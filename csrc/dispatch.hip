// dispatch.hip — host-side launch layer for the CDNA4 SGEMM kernel family.
// Reference-parity role: the if/else kernel dispatch of
// /root/reference/kernel/ft_sgemm/sgemm.cu:110-199, driven by the generated
// tier table; each tier's kernels live in their own translation unit
// (csrc/generated/kernel_<tier>.hip).

#include <hip/hip_runtime.h>

#include "ft_core.h"
#include "generated/tile_params.h"

namespace ftsgemm {

#define FT_DECL(name, BM, BN, BK, WM, WN, MM)                               \
  hipError_t launch_tier_##name(bool abft, bool inject, int M, int N, int K, \
                                const float* A, const float* B, float* C,    \
                                float alpha, float beta, float tau,          \
                                float inj_mag, int verify_windows, float* ws,\
                                hipStream_t stream);                         \
  size_t abft_ws_floats_##name(int M, int N, int K);
FT_TIER_LIST(FT_DECL)
#undef FT_DECL

hipError_t sgemm_tier_launch(int tier, bool abft, bool inject, int M, int N,
                             int K, const float* A, const float* B, float* C,
                             float alpha, float beta, float tau,
                             float inj_mag, int verify_windows, float* ws,
                             hipStream_t stream) {
  switch (tier) {
#define FT_CASE(name, BM, BN, BK, WM, WN, MM)                              \
  case FT_TIER_ID_##name:                                                  \
    return launch_tier_##name(abft, inject, M, N, K, A, B, C, alpha, beta, \
                              tau, inj_mag, verify_windows, ws, stream);
    FT_TIER_LIST(FT_CASE)
#undef FT_CASE
    default:
      return hipErrorInvalidValue;
  }
}

// Workspace floats needed by the fused-ABFT path of a tier (segment-sum
// matrices SA+SB); 0 for plain kernels.
size_t sgemm_abft_workspace_floats(int tier, int M, int N, int K) {
  switch (tier) {
#define FT_WS(name, BM, BN, BK, WM, WN, MM) \
  case FT_TIER_ID_##name:                   \
    return abft_ws_floats_##name(M, N, K);
    FT_TIER_LIST(FT_WS)
#undef FT_WS
    default:
      return 0;
  }
}

bool sgemm_tier_supported(int tier, int M, int N, int K) {
  switch (tier) {
#define FT_CHK(name, BM, BN, BK, WM, WN, MM)                               \
  case FT_TIER_ID_##name:                                                  \
    return !(M % BM || N % BN || K % BK || M % 4 || N % 4);
    FT_TIER_LIST(FT_CHK)
#undef FT_CHK
    default:
      return false;
  }
}

}  // namespace ftsgemm
