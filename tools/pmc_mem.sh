#!/usr/bin/env bash
# PMC evidence runs (VERDICT r01 weak #8 + next #9).  Counters-only —
# never combined with trace domains (pool rule).  TCC has 4 slots;
# FETCH_SIZE costs 3 and WRITE_SIZE 2, so they run in separate passes
# (MI355X_MICROARCH.md §rocprofv3 PMC slots).
#
# Pass set:
#   fetch/write x {kid 6 plain huge, kid 16 fused} at N=4096  -> HBM bytes
#   SQ wait/active x {classic, stream-K} plain huge at N=4608 -> tail-fill
# Writes a compact gpurun_out/pmc_mem_summary.txt (raw CSVs are deleted:
# they would blow the gpurun_out merge limit).
set -e
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
OUT=gpurun_out/pmc_mem
rm -rf "$OUT" && mkdir -p "$OUT"
SUM=gpurun_out/pmc_mem_summary.txt
: > "$SUM"

run_pmc() { # name counters kid size env...
  local name=$1 counters=$2 kid=$3 size=$4; shift 4
  env "$@" rocprofv3 --pmc $counters --output-format csv -d "$OUT/$name" \
    -o "$name" -- python tools/profile_one.py --kid "$kid" --size "$size" \
    --reps 3 > "$OUT/$name.run.log" 2>&1 || echo "PASS $name FAILED" >> "$SUM"
}

run_pmc fetch_k6   FETCH_SIZE  6 4096
run_pmc write_k6   WRITE_SIZE  6 4096
run_pmc fetch_k16  FETCH_SIZE 16 4096
run_pmc write_k16  WRITE_SIZE 16 4096
SQC="SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY,SQ_WAVE_CYCLES"
run_pmc sq_classic "$SQC" 6 4608 FT_SGEMM_STREAMK=0
run_pmc sq_streamk "$SQC" 6 4608 FT_SGEMM_STREAMK=1

python3 - "$OUT" >> "$SUM" <<'EOF'
import csv, glob, sys, os
out = sys.argv[1]
print("PMC summary (rocprofv3 counter collection, 3 reps + 2 warmups each).")
print("FETCH/WRITE_SIZE unit = KiB per the gfx94x formula: hbm_bytes ~=")
print("(FETCH_SIZE + WRITE_SIZE) * 1024 (guide: FETCH_SIZE undercounts a")
print("wide stream by ~2x).  Ideal huge-tier A+B operand traffic at 4096 =")
print("A*(N/BN) + B*(M/BM) = 64MiB*32 + 64MiB*16 = 3.0 GiB per GEMM + C rw.")
print()
for f in sorted(glob.glob(os.path.join(out, '**', '*.csv'), recursive=True)):
    tag = os.path.relpath(f, out).split(os.sep)[0]
    agg = {}
    with open(f) as fh:
        rd = csv.DictReader(fh)
        for row in rd:
            lk = {k.lower(): v for k, v in row.items()}
            name = lk.get('counter_name') or lk.get('counter-name')
            val = lk.get('counter_value') or lk.get('counter-value') or 0
            kern = (lk.get('kernel_name') or lk.get('kernel-name') or '')[:44]
            if not name:
                continue
            key = (kern, name)
            agg[key] = agg.get(key, 0.0) + float(val)
    for (kern, name), v in sorted(agg.items()):
        print(f"{tag:12s} {name:20s} {v:18.0f}  {kern}")
    print()
EOF
rm -rf "$OUT"
cat "$SUM"
