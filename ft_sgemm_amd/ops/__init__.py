"""Device ops: loads the in-tree HIP extension and exposes the kernel-id
dispatch surface (reference parity: sgemm.cu:110-199).

On a GPU box a missing/unbuilt extension is a hard error (no silent eager
fallback); on CPU-only hosts the golden model in .golden is the reference
implementation and the extension is not required.
"""

from __future__ import annotations

import torch

from ..kernel_table import (ERR_BOUND, ERROR_INJECT, KERNEL_TABLE, TIERS)
from . import golden  # noqa: F401

_C = None
_load_err = None
try:
    from .. import _C as _C  # type: ignore
except ImportError as e:  # extension not built
    _load_err = e


def have_extension() -> bool:
    return _C is not None


def _require_ext():
    if _C is None:
        raise RuntimeError(
            "ft_sgemm_amd._C HIP extension is not built. Run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(import error: {_load_err})")
    return _C


def tier_index(tier: str) -> int:
    return TIERS.index(tier)


def sgemm(tier: str, a: torch.Tensor, b: torch.Tensor, c: torch.Tensor,
          alpha: float = 1.0, beta: float = 0.0) -> torch.Tensor:
    """Plain hand-tiled MFMA SGEMM.  a:(K,M) b:(K,N) c:(N,M) fp32 CUDA
    tensors holding column-major A (MxK), B (NxK), C (MxN).  In-place on c."""
    _require_ext().sgemm(tier_index(tier), False, False, a, b, c,
                         alpha, beta, 0.0, 0.0, 20)
    return c


def ft_sgemm(tier: str, a: torch.Tensor, b: torch.Tensor, c: torch.Tensor,
             alpha: float = 1.0, beta: float = 0.0, inject: bool = True,
             tau: float = ERR_BOUND, inj_mag: float = ERROR_INJECT,
             verify_windows: int = 20) -> torch.Tensor:
    """Fused-ABFT MFMA SGEMM with in-kernel verify/locate/correct.  The
    default inject=True preserves the reference's always-self-testing
    property."""
    _require_ext().sgemm(tier_index(tier), True, inject, a, b, c,
                         alpha, beta, tau, inj_mag, verify_windows)
    return c


def rocblas_sgemm(a, b, c, alpha: float = 1.0, beta: float = 0.0):
    """Kernel id 0: the vendor-BLAS oracle."""
    _require_ext().rocblas_sgemm(a, b, c, alpha, beta)
    return c


def baseline_ft(a, b, c, alpha: float = 1.0, beta: float = 0.0,
                panel_k: int = 1024):
    """Kernel id 10: non-fused rocBLAS ABFT chain.  Returns (c, verdicts)."""
    res = _require_ext().baseline_ft(a, b, c, alpha, beta, panel_k)
    return c, res


def choose_tier(m: int, n: int, k: int):
    """Pick the fastest applicable tier for a problem shape, or None when no
    hand-tiled tier divides the shape (callers fall back to rocBLAS).

    Thresholds from the final measured 1024..6144 sweep
    (profiles/cli_sweep_r2.log, stream-K-enabled kernels): the huge
    256x128 macro-tile wins from ~384 tiles up (its stream-K twin covers
    the tail-waste sizes), the large 64x64 tier wins the 2048-3072 band
    (>=1024 blocks), and below that the medium tier's dense grid wins."""
    from .. import kernel_table as kt

    def fits(tier):
        t = kt.TILING[tier]
        return m % t["bm"] == 0 and n % t["bn"] == 0 and k % t["bk"] == 0

    if fits("huge") and (m // 256) * (n // 128) >= 384:
        return "huge"
    skinny = ("tall" if m >= 4 * n else "wide" if n >= 4 * m else None)
    if skinny and fits(skinny):
        return skinny
    if fits("large") and (m // 64) * (n // 64) >= 1024:
        return "large"
    for tier in ("medium", "large", "small"):
        if fits(tier):
            return tier
    return None


def sgemm_auto(a, b, c, alpha: float = 1.0, beta: float = 0.0):
    """Plain SGEMM with automatic tier selection; shapes no hand-written
    tier divides (e.g. M=100) fall back to the rocBLAS vendor path instead
    of raising."""
    k, m = a.shape
    n = b.shape[1]
    tier = choose_tier(m, n, k)
    if tier is None:
        return rocblas_sgemm(a, b, c, alpha, beta)
    return sgemm(tier, a, b, c, alpha, beta)


def ft_sgemm_auto(a, b, c, alpha: float = 1.0, beta: float = 0.0,
                  inject: bool = True):
    """Fused-ABFT SGEMM with automatic tier selection.  Shapes no tier
    divides fall back to the rocBLAS GEMM wrapped in the OFFLINE checksum
    ABFT chain (kernel id 10 semantics: detection verdicts, no in-kernel
    correction — the strongest FT available without a tile fit); a verdict
    above the threshold raises, making silent corruption impossible on the
    fallback path."""
    k, m = a.shape
    n = b.shape[1]
    tier = choose_tier(m, n, k)
    if tier is None:
        pk = k if k <= 1024 or k % 1024 else 1024
        _, (res_row, res_col) = baseline_ft(a, b, c, alpha, beta, panel_k=pk)
        # Squared residual l2 norms; fault-free roundoff at these operand
        # scales is orders of magnitude below ERR_BOUND^2.
        if max(res_row, res_col) > ERR_BOUND * ERR_BOUND:
            raise RuntimeError(
                f"ft_sgemm_auto fallback: offline ABFT verdict tripped "
                f"(res_row={res_row:.3e}, res_col={res_col:.3e}) — "
                f"uncorrectable fault in the rocBLAS fallback GEMM")
        return c
    return ft_sgemm(tier, a, b, c, alpha, beta, inject=inject)


def run_kernel_id(kid: int, a, b, c, alpha: float = 1.0, beta: float = 0.0,
                  inject: bool = True):
    """Dispatch by reference kernel id (0=rocBLAS, 1-6 plain tiers,
    7-9 rocBLAS fallback, 10 baseline, 11-16 fused-ABFT tiers)."""
    if kid in (0, 7, 8, 9):
        return rocblas_sgemm(a, b, c, alpha, beta)
    if kid == 10:
        return baseline_ft(a, b, c, alpha, beta)[0]
    name, tier, fused = KERNEL_TABLE[kid]
    if fused:
        return ft_sgemm(tier, a, b, c, alpha, beta, inject=inject)
    return sgemm(tier, a, b, c, alpha, beta)


def make_operands(m: int, n: int, k: int, device="cuda", seed: int = 10):
    """Column-major operands as (K,M)/(K,N)/(N,M) contiguous tensors with
    the reference value distribution (uniform (-0.9, 0.9), utils.cu:23)."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    a = (torch.rand((k, m), generator=g) * 1.8 - 0.9).to(device)
    b = (torch.rand((k, n), generator=g) * 1.8 - 0.9).to(device)
    c = torch.zeros((n, m), device=device)
    return a, b, c


def torch_reference(a: torch.Tensor, b: torch.Tensor, c: torch.Tensor,
                    alpha: float = 1.0, beta: float = 0.0) -> torch.Tensor:
    """Plain fp32 PyTorch reference of the same op: c' = alpha*(b^T a) + beta*c
    (in our storage convention this IS C = alpha*A B^T + beta*C)."""
    return alpha * (b.transpose(0, 1) @ a) + beta * c
