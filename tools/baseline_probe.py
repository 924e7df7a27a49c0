#!/usr/bin/env python3
"""Overhead A/B of the non-fused ABFT baseline (kernel id 10) vs rocBLAS
(reference ratio is 30.1% overhead at N=4096).

Modes: fast (custom reduction kernels) at verify_every 1 and 2, and the
strict rocBLAS-only "chain" mode; panel_k sweep via FT_SGEMM_PANEL_K.
Writes gpurun_out/baseline_probe.log
"""

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
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < 0.25:
        fn()
        torch.cuda.synchronize()
    reps = max(reps, int(200e9 / (2.0 * n * n * n)))
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
    sizes = [1024, 2048, 4096, 6144]
    log(f"{'size':>6} {'mode':>16} {'gflops':>9} {'rocblas':>9} {'ovh%':>6}")
    for n in sizes:
        a, b, c = ops.make_operands(n, n, n)
        rb = time_gflops(lambda: ops.rocblas_sgemm(a, b, c, 1.0, -1.5), n)
        for mode, env in [
            ("fast_j1", {"FT_SGEMM_VERIFY_EVERY": "1"}),
            ("fast_j2", {"FT_SGEMM_VERIFY_EVERY": "2"}),
            ("chain_j1", {"FT_SGEMM_BASELINE_MODE": "chain",
                          "FT_SGEMM_VERIFY_EVERY": "1"}),
        ]:
            for k, v in env.items():
                os.environ[k] = v
            pk = 1024 if n % 1024 == 0 else n
            g = time_gflops(
                lambda: ops.baseline_ft(a, b, c, 1.0, -1.5, panel_k=pk), n)
            for k in env:
                os.environ.pop(k, None)
            log(f"{n:>6} {mode:>16} {g:9.0f} {rb:9.0f} "
                f"{100 * (1 - g / rb):6.1f}")
        # verdict sanity on the last config
        _, (r0, r1) = ops.baseline_ft(a, b, c, 1.0, -1.5, panel_k=pk)
        assert r0 < 1.0 and r1 < 1.0, (r0, r1)
        del a, b, c
        torch.cuda.empty_cache()
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/baseline_probe.log", "w") as f:
        f.write("\n".join(LINES) + "\n")


if __name__ == "__main__":
    main()
