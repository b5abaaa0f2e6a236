#!/bin/bash
# Kernel perf sweep (the reference's paper/kernel/gpu/scripts/sweep.sh
# analog): table sizes x batch sizes, fused strategy, entry_size 1,
# one output file per config under benchmarks/sweep_out/.
# Usage: bash benchmarks/sweep.sh [PRF] [STRATEGY]
set -u
PRF=${1:-AES128}
STRATEGY=${2:-fused}
OUT=benchmarks/sweep_out
mkdir -p "$OUT"
for n in 8192 16384 32768 65536 131072 262144 524288 1048576 4194304 16777216; do
  for batch in 8 16 32 64 128 256 512 1024 2048 4096; do
    f="$OUT/${STRATEGY}_${PRF}_n${n}_b${batch}.txt"
    echo "== n=$n batch=$batch -> $f"
    python benchmarks/kernel_benchmark.py --strategy "$STRATEGY" --prf "$PRF" \
      --n "$n" --batch "$batch" --entry-size 1 --reps 5 > "$f" 2>&1
  done
done
python benchmarks/scrape.py "$OUT" > "$OUT/sweep.csv"
echo "wrote $OUT/sweep.csv"
