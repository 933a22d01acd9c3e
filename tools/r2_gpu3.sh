#!/bin/bash
# Round-2 GPU call #3: diff-profile the fused-residual regression (base vs
# flag under rocprofv3 --stats), re-run the ViT-g/14 rehearsal with the
# correct 98px local crops, and measure the input-path bench modes.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu3
mkdir -p "$OUT"

# 0. new GPU tests (pipeline end-to-end)
timeout 240 python -m pytest tests/test_data_gpu.py -q > "$OUT/pytest_data.log" 2>&1
echo "data_gpu=$?" >> "$OUT/summary.txt"

# 1. kernel-level diff: base vs DINOV3_FUSED_RESIDUAL=1 (trace/stats only)
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/$OUT/prof_base" -o base --output-format csv -- \
    python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 \
    > "$GRAFT_REPO_ROOT/$OUT/prof_base.log" 2>&1
echo "prof_base=$?" >> "$GRAFT_REPO_ROOT/$OUT/summary.txt"
DINOV3_FUSED_RESIDUAL=1 timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/$OUT/prof_fused" -o fused --output-format csv -- \
    python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 \
    > "$GRAFT_REPO_ROOT/$OUT/prof_fused.log" 2>&1
echo "prof_fused=$?" >> "$GRAFT_REPO_ROOT/$OUT/summary.txt"
cd "$GRAFT_REPO_ROOT"

# 2. ViT-g/14 occupancy rehearsal, local crops 98 (config #5)
timeout 420 python bench.py --arch vit_giant2 --patch-size 14 --grad-checkpointing \
    --steps 4 --warmup 2 --batch-size 64 --local-size 98 \
    > "$OUT/bench_vitg14.json" 2> "$OUT/bench_vitg14.err"
echo "vitg14=$?" >> "$OUT/summary.txt"

# 3. input-path bench modes (12 CPU workers; gpu-aug decouples from cores)
timeout 300 python bench.py --steps 10 --warmup 3 --data loader --num-workers 12 \
    > "$OUT/bench_loader.json" 2>> "$OUT/bench_modes.err"
echo "loader=$?" >> "$OUT/summary.txt"
timeout 300 python bench.py --steps 10 --warmup 3 --data gpu-aug --num-workers 6 \
    > "$OUT/bench_gpuaug.json" 2>> "$OUT/bench_modes.err"
echo "gpuaug=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
tail -3 "$OUT/pytest_data.log"
cat "$OUT/bench_vitg14.json"; tail -2 "$OUT/bench_vitg14.err"
tail -1 "$OUT/bench_loader.json"; tail -1 "$OUT/bench_gpuaug.json"
