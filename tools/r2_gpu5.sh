#!/bin/bash
# Round-2 GPU call #5: validate the sinkhorn chip-fill fix, the atomic-chain
# caps, and the bias-GELU ILP-8 variant; re-measure fused-residual.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu5
mkdir -p "$OUT"

# 1. full GPU suite (sinkhorn numerics, fused ops with the fixed test)
timeout 420 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"
DINOV3_FUSED_RESIDUAL=1 timeout 240 python -m pytest \
    tests/test_ops_gpu.py -q -k "fused_residual or ls_axpy_bias or ls_scatter" \
    > "$OUT/pytest_fused.log" 2>&1
echo "fused_suite=$?" >> "$OUT/summary.txt"

# 2. kernel micro-bench (sinkhorn after the fill fix)
timeout 240 python tools/bench_kernels.py --iters 30 > "$OUT/bench_kernels.log" 2>&1
echo "kernel_bench=$?" >> "$OUT/summary.txt"

# 3. step bench: base / base+ILP8 / fused / fused+ILP8
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_base.json"
DINOV3_BG_ILP8=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_ilp8.json"
DINOV3_FUSED_RESIDUAL=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_fused.json"
DINOV3_FUSED_RESIDUAL=1 DINOV3_BG_ILP8=1 timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench_fused_ilp8.json"

cat "$OUT/summary.txt"
tail -3 "$OUT/pytest.log"; tail -3 "$OUT/pytest_fused.log"
grep -E "sinkhorn|fmha" "$OUT/bench_kernels.log"
for f in "$OUT"/bench_*.json; do echo "$f"; cat "$f"; done
