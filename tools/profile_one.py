#!/usr/bin/env python3
"""Run one kernel id repeatedly for rocprofv3 capture.

    rocprofv3 --kernel-trace --stats -d gpurun_out/prof -- \
        python tools/profile_one.py --kid 16 --size 4096 --reps 10
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ft_sgemm_amd import ops  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--kid", type=int, default=16)
    ap.add_argument("--size", type=int, default=4096)
    ap.add_argument("--reps", type=int, default=10)
    ap.add_argument("--no-inject", action="store_true")
    ap.add_argument("--vw", type=int, default=20)
    args = ap.parse_args()
    n = args.size
    a, b, c = ops.make_operands(n, n, n)
    def run():
        if args.kid in range(11, 17):
            ops.ft_sgemm(["small","medium","large","tall","wide","huge"][args.kid-11],
                         a, b, c, 1.0, -1.5, inject=not args.no_inject,
                         verify_windows=args.vw)
        else:
            ops.run_kernel_id(args.kid, a, b, c, 1.0, -1.5,
                              inject=not args.no_inject)
    for _ in range(2):
        run()
    torch.cuda.synchronize()
    import time
    t0 = time.perf_counter()
    for _ in range(args.reps):
        run()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    gf = 2 * n**3 * args.reps / dt / 1e9
    print(f"kid={args.kid} size={n} reps={args.reps} "
          f"{dt/args.reps*1e3:.3f} ms/rep {gf:.0f} GFLOPS")


if __name__ == "__main__":
    main()
