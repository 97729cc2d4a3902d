#!/usr/bin/env bash
# One-command reproduction of every bench leg on a GPU box:
#   1. headline LUBM-2560 Q1-Q7 suite (bench.py default; includes the
#      parity gates, roofline probe, same-input cpu_baseline and a short
#      embedded emulator record in the same JSON line)
#   2. full emulator light-mix leg (1M queries, batched windows)
#   3. WatDiv-1B stress leg (20M products, star/linear/snowflake)
# Outputs one JSON line per leg under gpurun_out/ (or $OUT).
set -e
OUT=${OUT:-gpurun_out}
mkdir -p "$OUT"
cd "$(dirname "$0")/.."

echo "[bench_all] 1/3 LUBM-2560 suite" >&2
python bench.py > "$OUT/bench_suite.json" 2> "$OUT/bench_suite.log"

echo "[bench_all] 2/3 emulator light-mix" >&2
WK_SKIP_CPU_BASELINE=1 python bench.py --emu 1000000 \
    > "$OUT/bench_emulator.json" 2> "$OUT/bench_emulator.log"

echo "[bench_all] 3/3 WatDiv (20M products)" >&2
WK_SKIP_CPU_BASELINE=1 python bench.py --watdiv 20000000 --steps 400 \
    > "$OUT/bench_watdiv.json" 2> "$OUT/bench_watdiv.log"

echo "[bench_all] done:" >&2
for f in bench_suite bench_emulator bench_watdiv; do
    echo "== $f =="; cat "$OUT/$f.json"
done
