#!/bin/bash
# Round-2 GPU call #15: validate the hd-128 dkv split (numerics + high-res
# shape perf) and re-confirm the headline.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu15
mkdir -p "$OUT"

timeout 420 python -m pytest tests/test_fmha_rope_gpu.py tests/test_ops_gpu.py -q > "$OUT/pytest_fmha.log" 2>&1
echo "fmha_suite=$?" >> "$OUT/summary.txt"
timeout 240 python tools/bench_kernels.py --iters 30 > "$OUT/bench_kernels.log" 2>&1
echo "kernel_bench=$?" >> "$OUT/summary.txt"
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_default.json"

cat "$OUT/summary.txt"; tail -3 "$OUT/pytest_fmha.log"
grep -E "fmha" "$OUT/bench_kernels.log"
cat "$OUT/bench_default.json"
