#!/bin/bash
# Round-2 GPU call #8: validate the wave-per-row LN backward; profile the
# step to confirm LN/ls_scatter/bias_gelu wins and find what's left.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu18
mkdir -p "$OUT"

timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_default.json"

cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/$OUT/prof" -o def --output-format csv -- \
    python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 \
    > "$GRAFT_REPO_ROOT/$OUT/prof.log" 2>&1
echo "prof=$?" >> "$GRAFT_REPO_ROOT/$OUT/summary.txt"
cd "$GRAFT_REPO_ROOT"

cat "$OUT/summary.txt"
tail -3 "$OUT/pytest.log"
cat "$OUT/bench_default.json"
