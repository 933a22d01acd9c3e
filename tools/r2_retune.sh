#!/bin/bash
# Re-tune hipBLASLt algorithm selection with a larger search budget than the
# round-1 pass (longer per-shape duration + more iterations), then A/B.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_retune
mkdir -p "$OUT" "$OUT/csv"

# baseline with the shipped selections
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_shipped.json"

# tuning run: fresh file, generous budget
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME="$GRAFT_REPO_ROOT/$OUT/csv/tunableop.csv"
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=120
export PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=200
timeout 900 python bench.py --steps 3 --warmup 1 > "$OUT/tuning.log" 2>&1
echo "tuning=$?" >> "$OUT/summary.txt"
unset PYTORCH_TUNABLEOP_TUNING

# bench with the NEW selections
export PYTORCH_TUNABLEOP_FILENAME="$GRAFT_REPO_ROOT/$OUT/csv/tunableop.csv"
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_retuned.json"
echo "bench=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
for f in "$OUT"/bench_*.json; do echo "$f"; cat "$f"; echo; done
wc -l "$OUT"/csv/* 2>/dev/null
