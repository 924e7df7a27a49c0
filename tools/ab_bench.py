#!/usr/bin/env python3
"""Within-probe interleaved A/B of kernel variants (GEMM noise ~3%:
cdna_hip_programming.md §5.4 rule 24 — N variants x M rounds, one process,
report median and min)."""

import argparse
import statistics
import sys
import time

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ft_sgemm_amd import ops  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=4096)
    ap.add_argument("--reps", type=int, default=5)
    ap.add_argument("--rounds", type=int, default=5)
    ap.add_argument("--tier", default="huge")
    args = ap.parse_args()
    n = args.size
    t = args.tier
    a, b, c = ops.make_operands(n, n, n)

    variants = {
        "rocblas": lambda: ops.rocblas_sgemm(a, b, c, 1.0, -1.5),
        "plain": lambda: ops.sgemm(t, a, b, c, 1.0, -1.5),
        "abft_vw1": lambda: ops.ft_sgemm(t, a, b, c, 1.0, -1.5, inject=False,
                                         verify_windows=1),
        "abft_vw20_noinj": lambda: ops.ft_sgemm(t, a, b, c, 1.0, -1.5,
                                                inject=False),
        "abft_vw20": lambda: ops.ft_sgemm(t, a, b, c, 1.0, -1.5, inject=True),
    }
    results = {k: [] for k in variants}
    for v in variants.values():
        v()
    torch.cuda.synchronize()
    for _ in range(args.rounds):
        for name, fn in variants.items():
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.reps):
                fn()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.reps
            results[name].append(2 * n**3 / dt / 1e9)
    for name, vals in results.items():
        print(f"{name:18s} med={statistics.median(vals):8.0f} "
              f"max={max(vals):8.0f} GFLOPS")


if __name__ == "__main__":
    main()
