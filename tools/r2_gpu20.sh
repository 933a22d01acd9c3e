#!/bin/bash
# Repeated headline benches (within-box variance), a 2500-iteration soak,
# and the full suite — evidence-strengthening pass.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu20
mkdir -p "$OUT"

timeout 480 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

for i in 1 2 3; do
  timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 > "$OUT/bench_$i.json"
done

timeout 700 python -m dinov3_amd.train.train \
    --config-file dinov3_amd/configs/train/vitl_im1k_lin834.yaml \
    --output-dir /tmp/soak --no-resume --max-iterations 2500 \
    train.batch_size_per_gpu=32 train.num_workers=8 train.OFFICIAL_EPOCH_LENGTH=100 \
    optim.warmup_epochs=1 evaluation.eval_period_iterations=0 checkpointing.period=0 \
    > "$OUT/soak.log" 2>&1
echo "soak=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"; tail -2 "$OUT/pytest.log"
for f in "$OUT"/bench_*.json; do cat "$f"; echo; done
grep -oE "total_loss: [0-9.]+ \([0-9.]+\)" "$OUT/soak.log" | sed -n '1p;$p'
