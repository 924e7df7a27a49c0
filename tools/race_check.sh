#!/usr/bin/env bash
# Race check (SURVEY.md §5 race-detection row / VERDICT r01 next #9):
# the FT_PARANOID build drains every async-staging site (glds vmcnt + LDS
# lgkm + barrier) immediately; per-thread arithmetic order is unchanged, so
# each kernel's verification output must be BIT-IDENTICAL to the normal
# build.  Any byte difference = a staging/synchronisation race.
#
# Covers kernels 1-6 (plain) and 11-16 (fused ABFT + injection) at a
# straggler-ish size, in both the classic and the forced stream-K launch.
# (id 10 is excluded: its baseline rowsum kernel combines column slices
# with f32 atomics, which is add-order nondeterministic by design.)
#
# Run on a GPU box: bash tools/race_check.sh
set -e
cd "$(dirname "$0")/.."
N=${1:-1280}
OUT=gpurun_out/race_check
rm -rf "$OUT" && mkdir -p "$OUT"/{norm,para}{0,1}
LOG=gpurun_out/race_check.log
: > "$LOG"

run() { # binary dumpdir skmode
  FT_SGEMM_DUMP="$2" FT_SGEMM_STREAMK="$3" "$1" "$N" 0 512 1 6 > /dev/null
  FT_SGEMM_DUMP="$2" FT_SGEMM_STREAMK="$3" "$1" "$N" 0 512 11 16 > /dev/null
}

fail=0
for sk in 0 1; do
  run bin/ft_sgemm         "$OUT/norm$sk" "$sk"
  run bin/ft_sgemm_paranoid "$OUT/para$sk" "$sk"
  for f in "$OUT/norm$sk"/k*.bin; do
    b=$(basename "$f")
    if cmp -s "$f" "$OUT/para$sk/$b"; then
      echo "OK  bit-identical sk=$sk $b (N=$N)" >> "$LOG"
    else
      echo "RACE DIFF sk=$sk $b (N=$N)" >> "$LOG"
      fail=1
    fi
  done
done
if [ "$fail" = 0 ]; then
  echo "race check PASSED: all kernels bit-identical normal vs paranoid" >> "$LOG"
else
  echo "race check FAILED" >> "$LOG"
fi
rm -rf "$OUT"  # ~150 MB of raw dumps would blow the gpurun_out merge limit
cat "$LOG"
exit $fail
