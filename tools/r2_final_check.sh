#!/bin/bash
# Round-2 final validation: full GPU suite, long-steps headline bench, and a
# 1000-iteration optimization soak (real lr ramp -> loss must move down).
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_final
mkdir -p "$OUT"

timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

timeout 300 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 > "$OUT/bench30.json"
echo "bench=$?" >> "$OUT/summary.txt"

# 1000-iteration soak; checkpoints go to box-/tmp (not merged back), the log
# (small) comes back
timeout 600 python -m dinov3_amd.train.train \
    --config-file dinov3_amd/configs/train/vitl_im1k_lin834.yaml \
    --output-dir /tmp/soak --no-resume --max-iterations 1000 \
    train.batch_size_per_gpu=32 train.num_workers=8 train.OFFICIAL_EPOCH_LENGTH=100 \
    optim.warmup_epochs=1 evaluation.eval_period_iterations=0 checkpointing.period=0 \
    > "$OUT/soak.log" 2>&1
echo "soak=$?" >> "$OUT/summary.txt"
grep -E "Train \[" "$OUT/soak.log" | awk 'NR % 10 == 1 || /990/' | tail -12 > "$OUT/soak_tail.txt"

python -c "import torch,sys; sys.path.insert(0,'.'); import __graft_entry__ as g; g.smoke()" > "$OUT/smoke.log" 2>&1
echo "smoke=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
tail -3 "$OUT/pytest.log"
cat "$OUT/bench30.json"
grep -oE "total_loss: [0-9.]+ \([0-9.]+\)" "$OUT/soak.log" | head -3
grep -oE "total_loss: [0-9.]+ \([0-9.]+\)" "$OUT/soak.log" | tail -3
tail -2 "$OUT/smoke.log"
