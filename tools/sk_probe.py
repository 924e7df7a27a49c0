#!/usr/bin/env python3
"""A/B the stream-K path against the classic grid at the sweep sizes
(VERDICT r01 next #3).  Prints GFLOPS per (size, tier, kernel, mode) plus
the rocBLAS row, hipEvent-timed like the CLI (5 reps, beta=-1.5).

Usage: python tools/sk_probe.py [--sizes 1024,1536,...] [--tiers huge,large]
Writes gpurun_out/sk_probe.log
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ft_sgemm_amd import ops  # noqa: E402

LINES = []


def log(msg):
    print(msg, flush=True)
    LINES.append(msg)


def time_gflops(fn, n, reps=5):
    # steady-state warm: run the exact config for >=0.25 s so the clock ramp
    # and allocator are out of the timed region (the r01 probe's single
    # warm launch let run order skew A/B by up to 10%)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < 0.25:
        fn()
        torch.cuda.synchronize()
    reps = max(reps, int(200e9 / (2.0 * n * n * n)))  # >=~2ms timed
    beg = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    beg.record()
    for _ in range(reps):
        fn()
    end.record()
    torch.cuda.synchronize()
    ms = beg.elapsed_time(end)
    return 2.0 * n * n * n * reps / (ms * 1e-3) / 1e9


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes", default="1024,1536,2048,2560,3072,3584,4096,"
                                       "4608,5120,5632,6144")
    ap.add_argument("--tiers", default="huge,large")
    ap.add_argument("--reps", type=int, default=5)
    args = ap.parse_args()
    sizes = [int(s) for s in args.sizes.split(",")]
    tiers = args.tiers.split(",")

    # clock warm
    a, b, c = ops.make_operands(4096, 4096, 4096)
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < 2.0:
        ops.rocblas_sgemm(a, b, c, 1.0, -1.5)
        torch.cuda.synchronize()
    del a, b, c
    torch.cuda.empty_cache()

    hdr = f"{'size':>6} {'kernel':>22} {'classic':>9} {'streamk':>9} {'auto':>9}"
    log(hdr)
    for n in sizes:
        a, b, c = ops.make_operands(n, n, n)
        rb = time_gflops(lambda: ops.rocblas_sgemm(a, b, c, 1.0, -1.5), n,
                         args.reps)
        log(f"{n:>6} {'rocblas':>22} {rb:9.0f}")
        for tier in tiers:
            bm = 256 if tier == "huge" else 64
            bn = 128 if tier == "huge" else 64
            if n % bm or n % bn or n % 64:
                continue
            for fused in (False, True):
                name = ("abft_" if fused else "plain_") + tier
                row = {}
                for mode in ("0", "1", "2"):
                    os.environ["FT_SGEMM_STREAMK"] = mode
                    if fused:
                        fn = lambda: ops.ft_sgemm(tier, a, b, c, 1.0, -1.5,
                                                  inject=True)
                    else:
                        fn = lambda: ops.sgemm(tier, a, b, c, 1.0, -1.5)
                    row[mode] = time_gflops(fn, n, args.reps)
                log(f"{n:>6} {name:>22} {row['0']:9.0f} {row['1']:9.0f} "
                    f"{row['2']:9.0f}")
        del a, b, c
        torch.cuda.empty_cache()
    os.environ.pop("FT_SGEMM_STREAMK", None)

    # one debug launch per (size, tier) so the log records whether the
    # auto heuristic engaged and with what G (stderr from the extension)
    os.environ["FT_SGEMM_SK_DEBUG"] = "1"
    for n in sizes:
        for tier in tiers:
            bm = 256 if tier == "huge" else 64
            bn = 128 if tier == "huge" else 64
            if n % bm or n % bn or n % 64:
                continue
            a, b, c = ops.make_operands(n, n, n)
            ops.sgemm(tier, a, b, c, 1.0, 0.0)
            torch.cuda.synchronize()
            del a, b, c
            torch.cuda.empty_cache()
    os.environ.pop("FT_SGEMM_SK_DEBUG", None)

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/sk_probe.log", "w") as f:
        f.write("\n".join(LINES) + "\n")


if __name__ == "__main__":
    main()
