#!/bin/bash
# CPU thread-scaling sweep (the reference's benchmark_multithread_dpf.sh
# analog).  Usage: bash benchmarks/cpu_thread_sweep.sh [N] [PRF]
set -u
N=${1:-16384}
PRF=${2:-AES128}
OUT=benchmarks/sweep_out
mkdir -p "$OUT"
for t in 1 2 4 8 16 32 64 100; do
  f="$OUT/cpu_${PRF}_n${N}_t${t}.txt"
  python benchmarks/cpu_benchmark.py --n "$N" --prf "$PRF" --threads "$t" \
    --batch 128 --reps 2 > "$f" 2>&1
  tail -1 "$f"
done
