#!/bin/bash
# Round-2 GPU call #12: robustness — 30-iteration ViT-L training soak with
# the current defaults (loss decreasing, no NaNs, checkpoint save), and a
# ViT-7b/16 single-GPU rehearsal (hd-128 FMHA + SwiGLU + RMSNorm + ckpt).
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu12
mkdir -p "$OUT"

# 30-iteration ViT-L soak through the REAL trainer (loader + prefetcher)
timeout 600 python -m dinov3_amd.train.train \
    --config-file dinov3_amd/configs/train/vitl_im1k_lin834.yaml \
    --output-dir "$OUT/soak" --no-resume --max-iterations 30 \
    train.batch_size_per_gpu=32 train.num_workers=8 crops.local_crops_number=8 \
    evaluation.eval_period_iterations=0 checkpointing.period=25 \
    > "$OUT/soak.log" 2>&1
echo "soak=$?" >> "$OUT/summary.txt"
grep -E "Train \[" "$OUT/soak.log" | tail -4

# ViT-7b/16: 6.7B params on one GPU, activation checkpointing, batch 8
timeout 600 python bench.py --arch vit_7b --patch-size 16 --grad-checkpointing \
    --steps 3 --warmup 1 --batch-size 8 --local-crops 4 \
    > "$OUT/bench_vit7b.json" 2> "$OUT/bench_vit7b.err"
echo "vit7b=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
tail -1 "$OUT/bench_vit7b.json" 2>/dev/null
grep peak "$OUT/bench_vit7b.err" 2>/dev/null
tail -3 "$OUT/bench_vit7b.err" 2>/dev/null
