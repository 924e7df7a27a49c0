#!/usr/bin/env python3
"""Render our measured sweep next to the reference's published table.

Usage: python tools/compare_reference.py [profiles/cli_sweep_full3.log]

The reference numbers are the T4-class table in BASELINE.md (hardware is
~30x slower than one MI355X, so only the RATIOS — % of vendor BLAS and
fused-ABFT overhead % — are comparable)."""

import re
import sys
import os

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

def _ref_table():
    """Parse the reference's published table out of BASELINE.md."""
    ref = {}
    for line in open(os.path.join(ROOT, "BASELINE.md")):
        m = re.match(r"\| ([a-z_0-9]+)[^|]*\|(.*)\|", line)
        if not m:
            continue
        name = m.group(1)
        cells = [c.strip() for c in m.group(2).split("|")]
        try:
            vals = [int(c) for c in cells]
        except ValueError:
            continue
        if len(vals) == 11 and name not in ref:
            ref[name] = vals
    return {k: v[6] for k, v in ref.items()}  # N=4096 column


REF_4096 = _ref_table()


def main():
    path = sys.argv[1] if len(sys.argv) > 1 else os.path.join(
        ROOT, "profiles", "cli_sweep_final2.log")
    rows = {}
    sizes = None
    for line in open(path):
        m = re.match(r"Matrix Size\|(.*)\|", line)
        if m:
            sizes = [int(x) for x in m.group(1).split("|")]
        m = re.match(r"([a-z_0-9]+)\|(.*)\|", line)
        if m:
            rows[m.group(1)] = [int(x) for x in m.group(2).split("|")]
    i4096 = sizes.index(4096)
    print(f"{'kernel':24s} {'MI355X@4096':>12s} {'T4-ref':>8s} "
          f"{'speedup':>8s} {'%vendor(ours)':>14s} {'%vendor(ref)':>13s}")
    ours_blas = rows["cublas"][i4096]
    ref_blas = REF_4096["cublas"]
    for name, refv in REF_4096.items():
        ours = rows[name][i4096]
        print(f"{name:24s} {ours:12d} {refv:8d} {ours / refv:7.1f}x "
              f"{100 * ours / ours_blas:13.1f}% {100 * refv / ref_blas:12.1f}%")
    print("\nfused-ABFT overhead vs same-tier plain @4096 "
          "(ours vs reference):")
    for tier in ("small", "medium", "large", "tall", "wide", "huge"):
        o = 100 * (1 - rows[f"abft_kernel_{tier}"][i4096] /
                   rows[f"kernel_sgemm_{tier}"][i4096])
        rf = 100 * (1 - REF_4096[f"abft_kernel_{tier}"] /
                    REF_4096[f"kernel_sgemm_{tier}"])
        flag = "BETTER" if o < rf else "worse"
        print(f"  {tier:8s} ours {o:5.1f}%  ref {rf:5.1f}%  {flag}")


if __name__ == "__main__":
    main()
