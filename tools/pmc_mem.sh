#!/usr/bin/env bash
# HBM/L2 traffic counters for the tile-choice claim (VERDICT r01 weak #8:
# "PMC evidence lacks memory-side counters").  TCC has 4 slots and
# FETCH_SIZE costs 3 / WRITE_SIZE costs 2 (MI355X_MICROARCH.md §rocprofv3
# PMC slots), so the two run in separate passes.  Counters-only runs — no
# trace domains mixed in (pool rule).
#
# Writes gpurun_out/pmc_mem/ per-pass CSVs + gpurun_out/pmc_mem_summary.txt
set -e
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
OUT=gpurun_out/pmc_mem
rm -rf "$OUT" && mkdir -p "$OUT"
SUM=gpurun_out/pmc_mem_summary.txt
: > "$SUM"

run_pmc() { # name counters kid extra
  rocprofv3 --pmc $2 -d "$OUT/$1" -o "$1" -- \
    python tools/profile_one.py --kid $3 --size 4096 --reps 3 $4 \
    > "$OUT/$1.run.log" 2>&1 || echo "PASS $1 FAILED" >> "$SUM"
}

for kid in 6 16; do
  run_pmc "fetch_k$kid" "FETCH_SIZE" $kid ""
  run_pmc "write_k$kid" "WRITE_SIZE" $kid ""
done

python3 - "$OUT" >> "$SUM" <<'EOF'
import csv, glob, sys, os
out = sys.argv[1]
print("HBM traffic per GEMM launch at N=4096 (TCC FETCH_SIZE/WRITE_SIZE,")
print("counter unit = 32B*32ch... reported as raw counter sums; ideal")
print("A+B operand bytes per huge-tier GEMM = A*(N/128) + B*(M/256)")
print("= 64MB*32 + 64MB*16 = 3.0 GiB before cache reuse; C rw = 128 MiB)")
for f in sorted(glob.glob(os.path.join(out, "*", "*counter_collection.csv"))):
    tag = f.split(os.sep)[-2]
    agg = {}
    with open(f) as fh:
        for row in csv.DictReader(fh):
            name = row.get("Counter_Name") or row.get("counter_name")
            val = float(row.get("Counter_Value") or row.get("counter_value") or 0)
            kern = (row.get("Kernel_Name") or row.get("kernel_name") or "")[:40]
            agg.setdefault((kern, name), 0.0)
            agg[(kern, name)] += val
    for (kern, name), v in sorted(agg.items()):
        print(f"{tag:14s} {name:12s} {v:16.0f}  {kern}")
EOF
cat "$SUM"
