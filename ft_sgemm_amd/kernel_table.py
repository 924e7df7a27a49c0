"""Kernel-id table and CDNA4 tiling specification.

Kernel ids are API-compatible with the reference dispatch table
(/root/reference/kernel/ft_sgemm/sgemm.cu:110-199 and :235-237):

  0        vendor BLAS (rocBLAS here, cuBLAS in the reference)
  1..6     plain SGEMM tiers  small / medium / large / tall / wide / huge
  7..9     unused -> fall back to rocBLAS (reference: cuBLAS fallback)
  10       non-fused ABFT baseline composed from rocBLAS calls
  11..16   fused-ABFT tiers   small / medium / large / tall / wide / huge

The tiling is re-derived for CDNA4 (gfx950): 64-lane wavefronts and MFMA
matrix cores, NOT the reference's 32-thread-warp register tiling
(/root/reference/kernel/ft_sgemm/code_gen/main.py:8-16 is the reference's
7-tuple table; ours is a different parameterisation because the per-thread
mr x nr register tile is replaced by per-wave MFMA fragments).

Fields per tier:
  bm, bn     block (workgroup) output tile
  bk         K-depth of one LDS panel (double buffered)
  bkf        K-depth for the FUSED-ABFT twin (defaults to bk).  The ABFT
             checksum strips add 1 KB/wave of LDS; on the skinny tiers
             whose plain kernel sits exactly at an LDS-occupancy boundary
             (tall/wide: 20 KB = 8 blocks/CU) that pushes them over a
             cliff (22 KB -> 7) — a smaller fused panel buys the
             occupancy back.
  wm, wn     per-wave output sub-tile (waves = (bm/wm)*(bn/wn))
  mfma       'f32_32x32x2' or 'f32_16x16x4' (f32-input MFMA shapes on gfx950)
"""

from collections import OrderedDict

# BK=16 on the mid tiers: measured +8..+19% plain and up to +23% fused at
# N=2048..4096 (tools/probe_pipeline.hip PROBE_ONLY=T): halving the LDS
# panel keeps one more block resident per CU, which covers the per-panel
# barrier park.
TILING = OrderedDict(
    small=dict(bm=16, bn=16, bk=32, wm=16, wn=16, mfma="f32_16x16x4"),
    medium=dict(bm=32, bn=32, bk=16, wm=32, wn=32, mfma="f32_32x32x2"),
    # large: 2 waves of 32x64 beat the single 64x64 wave by ~6% plain and
    # ~11% fused at N=4096 (probe_pipeline PROBE_ONLY=T)
    # large: bkf=8 (fused LDS 18 KB -> 10 KB: 8 -> 13+ blocks/CU)
    large=dict(bm=64, bn=64, bk=16, bkf=8, wm=32, wn=64,
               mfma="f32_32x32x2", streamk=True),
    # tall: 16x16x4 fragments (8 MFMAs vs 6 encode fmas per k-step) were
    # measured EQUAL to 32x32x2 (2 vs 2) — plain 109.3 vs 110.3, fused
    # 86.8 vs 87.0 at N=4096 — so the encode:MFMA issue ratio is NOT what
    # pins the tall fused overhead at ~21%; the r1-validated shape stays.
    # bkf=8: the fused twin's halved panel keeps 20 KB/block -> 8
    # blocks/CU (see bkf note above).
    # bkf=8 measured: tall fused overhead 21% -> 15-16% (encode cost
    # 16.5% -> 2.2%: the LDS-occupancy cliff, 22 KB -> 7 blocks/CU, was
    # the real tall bottleneck, not the encode's issue slots); wide
    # same-round A/B also favors bkf=8 (noinj 95.4k vs 84.3k GFLOPS at
    # 4096 — an earlier revert compared against a different box's r1 run)
    tall=dict(bm=128, bn=32, bk=16, bkf=8, wm=64, wn=32,
              mfma="f32_32x32x2"),
    wide=dict(bm=32, bn=128, bk=16, bkf=8, wm=32, wn=64,
              mfma="f32_32x32x2"),
    # 256x128 macro-tile, BK=16: measured 135 TF vs 128 TF for 128x128x32
    # at N=4096 (tools/probe_pipeline.hip T3/T6 vs PA) — bigger M-tile cuts
    # total A/B traffic 25% and the 48 KB LDS keeps 2 blocks/CU resident.
    # streamk: also build the stream-K twin (csrc/ft_streamk.hpp) — the
    # launcher auto-selects it at grid-straggler sizes where the classic
    # tile-per-workgroup grid leaves a tail dispatch round mostly idle.
    huge=dict(bm=256, bn=128, bk=16, wm=128, wn=64, mfma="f32_32x32x2",
              streamk=True),
)

TIERS = list(TILING.keys())


def threads(tier: str) -> int:
    t = TILING[tier]
    return 64 * (t["bm"] // t["wm"]) * (t["bn"] // t["wn"])


# id -> (name, tier or None, fused_abft)
KERNEL_TABLE = {0: ("cublas", None, False)}
for i, tier in enumerate(TIERS):
    KERNEL_TABLE[1 + i] = (f"kernel_sgemm_{tier}", tier, False)
KERNEL_TABLE[10] = ("abft_baseline", None, False)
for i, tier in enumerate(TIERS):
    KERNEL_TABLE[11 + i] = (f"abft_kernel_{tier}", tier, True)

# Names in the exact order of the reference perf sweep
# (/root/reference/kernel/ft_sgemm/sgemm.cu:235-237: ids {0,1..6,10,11..16}).
PERF_SWEEP_IDS = [0, 1, 2, 3, 4, 5, 6, 10, 11, 12, 13, 14, 15, 16]
KERNEL_NAMES = {k: v[0] for k, v in KERNEL_TABLE.items()}

# Fault-injection / ABFT constants (parity with the generated reference
# kernels, include_code_gen/ft_sgemm_huge.cuh:49-51: err_bound1=9500,
# error_inject=1e4, 20 injections per GEMM at period K/20).
ERR_BOUND = 9500.0
ERROR_INJECT = 10000.0
N_INJECT = 20
