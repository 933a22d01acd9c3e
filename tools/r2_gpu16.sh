#!/bin/bash
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu16
mkdir -p "$OUT"
# numerics: the rows variant must match the fp32 oracle
DINOV3_BG_ROWS=1 timeout 240 python -m pytest tests/test_ops_gpu.py -q -k "gelu" > "$OUT/pytest_rows.log" 2>&1
echo "rows_suite=$?" >> "$OUT/summary.txt"
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_default.json"
DINOV3_BG_ROWS=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_rows.json"
cat "$OUT/summary.txt"; tail -2 "$OUT/pytest_rows.log"
for f in "$OUT"/bench_*.json; do echo "$f"; cat "$f"; echo; done
