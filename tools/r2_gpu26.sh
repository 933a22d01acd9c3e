#!/bin/bash
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu26
mkdir -p "$OUT"
timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"
timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 > "$OUT/bench.json"
cat "$OUT/summary.txt"; tail -2 "$OUT/pytest.log"; cat "$OUT/bench.json"
