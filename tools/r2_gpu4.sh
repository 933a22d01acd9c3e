#!/bin/bash
# Round-2 GPU call #4: fused-residual kernel-level diff (fixed profiler
# flags) and ViT-g/14 operating-point sweep for config #5.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu4
mkdir -p "$OUT"

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

# ViT-g/14 operating points: no-checkpointing vs checkpointing at batch 128
timeout 420 python bench.py --arch vit_giant2 --patch-size 14 \
    --steps 4 --warmup 2 --batch-size 64 --local-size 98 \
    > "$OUT/vitg14_nockpt.json" 2> "$OUT/vitg14_nockpt.err"
echo "vitg_nockpt=$?" >> "$OUT/summary.txt"
timeout 420 python bench.py --arch vit_giant2 --patch-size 14 --grad-checkpointing \
    --steps 4 --warmup 2 --batch-size 128 --local-size 98 \
    > "$OUT/vitg14_b128.json" 2> "$OUT/vitg14_b128.err"
echo "vitg_b128=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
for f in "$OUT"/vitg14_*.json; do echo "$f"; tail -1 "$f"; done
grep peak "$OUT"/vitg14_*.err
ls "$OUT/prof_base" "$OUT/prof_fused" 2>/dev/null | head
