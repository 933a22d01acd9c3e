#!/bin/bash
# Round-2 opening GPU call: validate everything staged in round 1 and measure
# the gated optimizations, in one gpurun invocation. Usage:
#   /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/round2_validate.sh'
# Then read gpurun_out/r2_validate/* and flip the winning flags to default.
set -x
export TMPDIR=/tmp
OUT=gpurun_out/r2_validate
mkdir -p "$OUT"

# 1. baseline correctness: full GPU suite (ungated paths)
timeout 300 python -m pytest tests -m gpu -q > "$OUT/pytest_default.log" 2>&1
echo "default_suite=$?" >> "$OUT/summary.txt"

# 2. gated kernels: dkv64 + fused residual numerics
DINOV3_FMHA_DKV64=1 timeout 180 python -m pytest tests/test_fmha_rope_gpu.py -q \
    > "$OUT/pytest_dkv64.log" 2>&1
echo "dkv64_suite=$?" >> "$OUT/summary.txt"
DINOV3_FUSED_RESIDUAL=1 timeout 180 python -m pytest \
    tests/test_ops_gpu.py -q -k "fused_residual or ls_axpy_bias or ls_scatter" \
    > "$OUT/pytest_fused.log" 2>&1
echo "fused_suite=$?" >> "$OUT/summary.txt"

# 2b. per-kernel micro-benchmarks (HIP vs torch)
timeout 120 python tools/bench_kernels.py --iters 30 > "$OUT/bench_kernels.log" 2>&1
echo "kernel_bench=$?" >> "$OUT/summary.txt"

# 3. bench: baseline, then each flag, then both
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_base.json"
DINOV3_FMHA_DKV64=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null \
    | tail -1 > "$OUT/bench_dkv64.json"
DINOV3_FUSED_RESIDUAL=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null \
    | tail -1 > "$OUT/bench_fused.json"
DINOV3_FMHA_DKV64=1 DINOV3_FUSED_RESIDUAL=1 timeout 200 python bench.py --steps 15 --warmup 4 \
    2>/dev/null | tail -1 > "$OUT/bench_both.json"

cat "$OUT/summary.txt"
for f in "$OUT"/bench_*.json; do echo "$f"; cat "$f"; done
