#!/bin/bash
# Round-2 GPU call #7: validate shadow-copy atomics + hd-128 high-occupancy
# FMHA; remeasure the headline and the high-res shape.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu7
mkdir -p "$OUT"

timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

timeout 240 python tools/bench_kernels.py --iters 30 > "$OUT/bench_kernels.log" 2>&1
echo "kernel_bench=$?" >> "$OUT/summary.txt"

timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_default.json"

cat "$OUT/summary.txt"
tail -3 "$OUT/pytest.log"
grep -E "sinkhorn|fmha|ls_axpy|layernorm|bias_gelu" "$OUT/bench_kernels.log"
cat "$OUT/bench_default.json"
