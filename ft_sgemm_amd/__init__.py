"""ft_sgemm_amd — MI355X-native fault-tolerant SGEMM framework.

A from-scratch CDNA4 (gfx950) re-design of the capabilities of
shixun404/Fault-Tolerant-SGEMM-on-NVIDIA-GPUs (arXiv:2305.01024):

* six hand-tiled fp32 GEMM kernels (small/medium/large/tall/wide/huge)
  built on f32-input MFMA (`v_mfma_f32_32x32x2_f32` / `v_mfma_f32_16x16x4_f32`)
  with LDS double-buffered A/B panels,
* fused-ABFT twins that maintain wave-level column checksums (plain +
  row-index-weighted) of the output tile in registers from precomputed
  segment sums, periodically run a cheap whole-tile detect, locate a
  corrupted accumulator element by the residual ratio (column is
  lane-local, row = round(rw/rc)) and correct it in place,
* a non-fused ABFT baseline composed from rocBLAS calls,
* a rocBLAS oracle path (kernel id 0),
* multi-GPU scaling via torch.distributed over RCCL/xGMI
  (replicated weak scaling + block-row-distributed large-N SGEMM).

Matrix convention (identical to the reference CLI semantics,
/root/reference/kernel/ft_sgemm/sgemm.cu:108): C = alpha * A @ B^T + beta * C
with A (MxK), B (NxK) and C (MxN) all column-major.  In torch/numpy terms a
column-major MxK matrix is held as a row-major contiguous (K, M) tensor `a`
with a[k, i] = A[i, k].
"""

__version__ = "0.3.0"

from . import utils  # noqa: F401
from .kernel_table import KERNEL_TABLE, KERNEL_NAMES, TILING  # noqa: F401
