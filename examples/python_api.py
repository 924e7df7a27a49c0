#!/usr/bin/env python3
"""Library usage examples (run on a ROCm GPU box).

Tensor convention: a column-major MxK fp32 matrix is a contiguous (K, M)
CUDA tensor (same bytes, zero copies) — see ft_sgemm_amd/__init__.py.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ft_sgemm_amd import ops

# --- operands: C = alpha * A @ B^T + beta * C, A (MxK), B (NxK), C (MxN)
m = n = k = 4096
a, b, c = ops.make_operands(m, n, k)          # uniform (-0.9, 0.9), seed 10

# 1. plain hand-tiled MFMA SGEMM, explicit tier
ops.sgemm("huge", a, b, c, alpha=1.0, beta=0.0)

# 2. fused-ABFT SGEMM with the always-on 20-fault self-test: every fault is
#    detected, located (ratio locate) and corrected in-kernel; the result
#    still matches the clean product
ops.ft_sgemm("huge", a, b, c, alpha=1.0, beta=0.0, inject=True)

# 3. automatic tier selection (grid-fill aware)
ops.ft_sgemm_auto(a, b, c)

# 4. vendor-BLAS oracle and the non-fused rocBLAS ABFT baseline
ops.rocblas_sgemm(a, b, c)
c2, (res_row, res_col) = ops.baseline_ft(a, b, c, panel_k=1024)
print("baseline verdicts (squared residual norms):", res_row, res_col)

# 5. reference-id dispatch (0=rocBLAS, 1-6 plain, 10 baseline, 11-16 fused)
ops.run_kernel_id(16, a, b, c)

# 5b. stream-K control: the launcher auto-selects the stream-K twin at
#     grid-straggler sizes (measured gates); force / disable explicitly:
os.environ["FT_SGEMM_STREAMK"] = "1"   # force (pipelined callers win from
ops.sgemm("huge", a, b, c)             # N>=1024 up); "0" disables; unset
os.environ.pop("FT_SGEMM_STREAMK")     # = auto

# 5c. odd shapes fall back to rocBLAS (FT entry adds the offline ABFT
#     verdict chain) instead of raising
ao, bo, co = ops.make_operands(100, 257, 65)
ops.ft_sgemm_auto(ao, bo, co)

# 6. distributed block-row SGEMM (one process per GPU over RCCL; see
#    bench.py --mode blockrow for the full multi-rank setup)
from ft_sgemm_amd.parallel import block_row_sgemm

block_row_sgemm(a, b, c, panel_k=1024,
                gemm_fn=lambda ap, bp, cl, al, be:
                    ops.ft_sgemm("huge", ap, bp, cl, al, be, inject=True))

torch.cuda.synchronize()
print("examples OK")
