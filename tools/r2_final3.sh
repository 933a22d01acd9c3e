#!/bin/bash
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_final3
mkdir -p "$OUT"
timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"
timeout 420 python -m dinov3_amd.train.train \
    --config-file dinov3_amd/configs/train/vits_smoke.yaml \
    --output-dir /tmp/cnx --no-resume --max-iterations 3 \
    student.arch=convnext_tiny ibot.loss_weight=0 compute_precision.param_dtype=bf16 \
    train.batch_size_per_gpu=16 crops.global_crops_size=224 crops.local_crops_size=96 \
    crops.local_crops_number=4 checkpointing.period=0 > "$OUT/convnext_train.log" 2>&1
echo "convnext_train=$?" >> "$OUT/summary.txt"
timeout 200 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1 > "$OUT/bench.json"
cat "$OUT/summary.txt"; tail -2 "$OUT/pytest.log"
grep -oE "total_loss: [0-9.]+" "$OUT/convnext_train.log" | tail -1
cat "$OUT/bench.json"
