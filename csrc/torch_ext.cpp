// torch_ext.cpp — PyTorch-ROCm bindings for the CDNA4 FT-SGEMM kernel
// family.  Tensor convention: a column-major MxK fp32 matrix is passed as a
// contiguous row-major (K, M) CUDA tensor (same bytes, zero copies) —
// see ft_sgemm_amd/__init__.py.

#include <ATen/cuda/CUDAContext.h>
#include <torch/extension.h>

#include "ft_core.h"

namespace {

void check_inputs(const at::Tensor& a, const at::Tensor& b,
                  const at::Tensor& c, int64_t& M, int64_t& N, int64_t& K) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda() && c.is_cuda(),
              "ft_sgemm: tensors must be on a GPU device");
  TORCH_CHECK(a.scalar_type() == at::kFloat &&
                  b.scalar_type() == at::kFloat &&
                  c.scalar_type() == at::kFloat,
              "ft_sgemm: fp32 only (SGEMM)");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous() && c.is_contiguous(),
              "ft_sgemm: tensors must be contiguous");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && c.dim() == 2,
              "ft_sgemm: 2-D tensors expected");
  K = a.size(0);
  M = a.size(1);
  N = b.size(1);
  TORCH_CHECK(b.size(0) == K, "ft_sgemm: A is (K,M) and B must be (K,N)");
  TORCH_CHECK(c.size(0) == N && c.size(1) == M,
              "ft_sgemm: C must be (N,M) for column-major MxN");
}

void sgemm(int64_t tier, bool abft, bool inject, at::Tensor a, at::Tensor b,
           at::Tensor c, double alpha, double beta, double tau,
           double inj_mag, int64_t verify_windows) {
  int64_t M, N, K;
  check_inputs(a, b, c, M, N, K);
  TORCH_CHECK(ftsgemm::sgemm_tier_supported((int)tier, M, N, K),
              "ft_sgemm: tier ", tier, " requires M,N,K multiples of its "
              "tile (M=", M, " N=", N, " K=", K, ")");
  auto stream = at::cuda::getCurrentCUDAStream();
  at::Tensor ws;
  float* ws_ptr = nullptr;
  if (abft) {
    // segment-checksum scratch (SA/SB), via the caching allocator
    size_t n = ftsgemm::sgemm_abft_workspace_floats((int)tier, (int)M,
                                                    (int)N, (int)K);
    ws = at::empty({(int64_t)n}, a.options());
    ws_ptr = ws.mutable_data_ptr<float>();
  }
  hipError_t err = ftsgemm::sgemm_tier_launch(
      (int)tier, abft, inject, (int)M, (int)N, (int)K,
      a.const_data_ptr<float>(), b.const_data_ptr<float>(),
      c.mutable_data_ptr<float>(), (float)alpha, (float)beta, (float)tau,
      (float)inj_mag, (int)verify_windows, ws_ptr, stream.stream());
  TORCH_CHECK(err == hipSuccess,
              "ft_sgemm launch failed: ", hipGetErrorString(err));
}

void rocblas_sgemm_nt(at::Tensor a, at::Tensor b, at::Tensor c, double alpha,
                      double beta) {
  int64_t M, N, K;
  check_inputs(a, b, c, M, N, K);
  auto stream = at::cuda::getCurrentCUDAStream();
  int st = ftsgemm::rocblas_sgemm_nt(
      (int)M, (int)N, (int)K, a.const_data_ptr<float>(),
      b.const_data_ptr<float>(), c.mutable_data_ptr<float>(), (float)alpha,
      (float)beta, stream.stream());
  TORCH_CHECK(st == 0, "rocblas_sgemm failed with status ", st);
}

std::tuple<double, double> baseline_ft(at::Tensor a, at::Tensor b,
                                       at::Tensor c, double alpha,
                                       double beta, int64_t panel_k) {
  int64_t M, N, K;
  check_inputs(a, b, c, M, N, K);
  auto opts = a.options();
  int64_t mx = std::max(M, N);
  at::Tensor ones = at::ones({mx}, opts);
  at::Tensor row_c = at::empty({M}, opts), col_c = at::empty({N}, opts);
  at::Tensor s_a = at::empty({panel_k}, opts), s_b = at::empty({panel_k}, opts);
  at::Tensor ref_row = at::empty({M}, opts), ref_col = at::empty({N}, opts);
  // one verdict slot pair per (potentially) verified panel (worst-panel
  // semantics)
  int64_t npanels = (K + panel_k - 1) / panel_k;
  at::Tensor d_res = at::empty({2 * npanels}, opts);
  ftsgemm::BaselineWorkspace ws{
      ones.mutable_data_ptr<float>(),    row_c.mutable_data_ptr<float>(),
      col_c.mutable_data_ptr<float>(),   s_a.mutable_data_ptr<float>(),
      s_b.mutable_data_ptr<float>(),     ref_row.mutable_data_ptr<float>(),
      ref_col.mutable_data_ptr<float>(), d_res.mutable_data_ptr<float>()};
  float res_row = 0.f, res_col = 0.f;
  auto stream = at::cuda::getCurrentCUDAStream();
  int st = ftsgemm::baseline_ft_sgemm(
      (int)M, (int)N, (int)K, a.const_data_ptr<float>(),
      b.const_data_ptr<float>(), c.mutable_data_ptr<float>(), (float)alpha,
      (float)beta, ws, (int)panel_k, &res_row, &res_col, stream.stream());
  TORCH_CHECK(st == 0, "baseline_ft_sgemm failed with status ", st);
  return {(double)res_row, (double)res_col};
}

bool tier_supported(int64_t tier, int64_t M, int64_t N, int64_t K) {
  return ftsgemm::sgemm_tier_supported((int)tier, (int)M, (int)N, (int)K);
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sgemm", &sgemm, "tiered MFMA SGEMM (column-major C=aAB^T+bC)");
  m.def("rocblas_sgemm", &rocblas_sgemm_nt, "rocBLAS oracle SGEMM");
  m.def("baseline_ft", &baseline_ft, "non-fused rocBLAS ABFT baseline");
  m.def("tier_supported", &tier_supported, "tile divisibility check");
}
