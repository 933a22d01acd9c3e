#!/bin/bash
# Comprehensive end-of-round validation on one box: full suite, repeated
# headline benches, ConvNeXt-student GPU train, multi-resolution GPU train,
# input-path benches, smoke.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_final2
mkdir -p "$OUT"

timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

for i in 1 2; do
  timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 > "$OUT/bench_$i.json"
done
timeout 300 python bench.py --steps 10 --warmup 3 --data loader --num-workers 12 2>/dev/null | tail -1 > "$OUT/bench_loader.json"

# ConvNeXt DINO-only student on device, 3 iterations through the trainer
timeout 420 python -m dinov3_amd.train.train \
    --config-file dinov3_amd/configs/train/vits_smoke.yaml \
    --output-dir /tmp/cnx --no-resume --max-iterations 3 \
    student.arch=convnext_tiny ibot.loss_weight=0 compute_precision.param_dtype=bf16 \
    train.batch_size_per_gpu=16 crops.global_crops_size=224 crops.local_crops_size=96 \
    crops.local_crops_number=4 checkpointing.period=0 > "$OUT/convnext_train.log" 2>&1
echo "convnext_train=$?" >> "$OUT/summary.txt"

# multi-resolution schedule on device
timeout 420 python -m dinov3_amd.train.train \
    --config-file dinov3_amd/configs/train/vitl_im1k_lin834.yaml \
    --output-dir /tmp/mres --no-resume --max-iterations 4 \
    "crops.global_crops_size=[224,160]" "crops.local_crops_size=[96,64]" \
    "crops.global_local_crop_pairs_ratios=[0.5,0.5]" \
    train.batch_size_per_gpu=16 train.num_workers=6 \
    evaluation.eval_period_iterations=0 checkpointing.period=0 \
    > "$OUT/mres_train.log" 2>&1
echo "mres_train=$?" >> "$OUT/summary.txt"

python -c "import torch,sys; sys.path.insert(0,'.'); import __graft_entry__ as g; g.smoke()" > "$OUT/smoke.log" 2>&1
echo "smoke=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
tail -2 "$OUT/pytest.log"
for f in "$OUT"/bench_*.json; do cat "$f"; echo; done
grep -oE "total_loss: [0-9.]+" "$OUT/convnext_train.log" | tail -1
grep -oE "total_loss: [0-9.]+" "$OUT/mres_train.log" | tail -1
tail -1 "$OUT/smoke.log"
