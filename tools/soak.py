#!/usr/bin/env python3
"""Randomized soak test: random shapes x tiers x seeds x alpha/beta x
injection settings, fused-ABFT vs the plain PyTorch fp32 reference.

    python tools/soak.py --trials 100 [--seed 0]

Exits non-zero on the first failure with a full repro line.
"""

import argparse
import os
import random
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ft_sgemm_amd import ops  # noqa: E402
from ft_sgemm_amd.kernel_table import TILING  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=100)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--max-dim", type=int, default=2048)
    args = ap.parse_args()
    rng = random.Random(args.seed)
    fails = 0
    for t in range(args.trials):
        tier = rng.choice(list(TILING))
        spec = TILING[tier]
        m = spec["bm"] * rng.randint(1, max(1, args.max_dim // spec["bm"]))
        n = spec["bn"] * rng.randint(1, max(1, args.max_dim // spec["bn"]))
        k = spec["bk"] * rng.randint(1, max(1, args.max_dim // spec["bk"]))
        alpha = rng.choice([1.0, 0.5, -2.0])
        beta = rng.choice([0.0, 1.0, -1.5])
        inject = rng.random() < 0.7
        vw = rng.choice([1, 5, 20, 40])
        seed = rng.randint(0, 10_000)
        op = rng.choice(["ft", "ft", "ft", "plain", "baseline", "blockrow",
                         "auto"])
        a, b, c = ops.make_operands(m, n, k, seed=seed)
        if beta != 0.0:
            c.normal_(0, 1.0)
        ref = ops.torch_reference(a, b, c, alpha, beta)
        if op == "ft":
            ops.ft_sgemm(tier, a, b, c, alpha, beta, inject=inject,
                         verify_windows=vw)
        elif op == "plain":
            ops.sgemm(tier, a, b, c, alpha, beta)
        elif op == "baseline":
            pk = rng.choice([256, 512, 1024])
            ops.baseline_ft(a, b, c, alpha, beta, panel_k=pk)
        elif op == "blockrow":
            from ft_sgemm_amd.parallel import block_row_sgemm
            # realistic panel widths (bench.py uses >= 1024): with tiny
            # panels + inject + verify_windows=1 the deterministic injector
            # hits the SAME output element in every panel-GEMM, and each
            # correction's ~1e-4 checksum-roundoff residue accumulates
            # linearly (35 panels * 2e-4 * |alpha| can cross the 1e-2
            # tolerance).  That accumulation is inherent to repeated
            # same-site ABFT corrections, not a defect.
            pk = k
            for cand in (1024, 512, 256):
                if k % cand == 0:
                    pk = cand
                    break
            block_row_sgemm(
                a, b, c, panel_k=pk, alpha=alpha, beta=beta,
                gemm_fn=lambda ap, bp, cl, al, be: ops.ft_sgemm(
                    tier, ap, bp, cl, al, be, inject=inject))
        else:
            ops.ft_sgemm_auto(a, b, c, alpha, beta, inject=inject)
        torch.cuda.synchronize()
        diff = (ref - c).abs()
        rel = diff / ref.abs().clamp_min(1e-30)
        bad = int(((diff > 1e-2) & (rel > 1e-2)).sum())
        line = (f"trial {t}: op={op} tier={tier} m={m} n={n} k={k} alpha={alpha} "
                f"beta={beta} inject={inject} vw={vw} seed={seed} "
                f"bad={bad} maxdiff={diff.max().item():.2e}")
        if bad:
            print("FAIL", line)
            fails += 1
        elif t % 20 == 0:
            print("ok  ", line)
    print(f"soak: {args.trials} trials, {fails} failures")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
