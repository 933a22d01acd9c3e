#!/bin/bash
# Round-2 GPU call #11: validate the RR revert + H2D prefetcher; A/B the
# leftover gated variants under the current build.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu11
mkdir -p "$OUT"

timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_default.json"
DINOV3_BG_ILP8=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_ilp8.json"
DINOV3_FMHA_DKV64=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_dkv64.json"
timeout 300 python bench.py --steps 10 --warmup 3 --data loader --num-workers 12 2>/dev/null | tail -1 > "$OUT/bench_loader.json"

cat "$OUT/summary.txt"; tail -3 "$OUT/pytest.log"
for f in "$OUT"/bench_*.json; do echo "$f"; cat "$f"; echo; done
