#!/bin/bash
# No-risk evidence pass: longer ViT-g/14 bench, ViT-7b @512px high-res-adapt
# first-stage rehearsal (hd-128 long-N in a real model), giant2 profile.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_evidence
mkdir -p "$OUT"

timeout 420 python bench.py --arch vit_giant2 --patch-size 14 \
    --steps 8 --warmup 2 --batch-size 64 --local-size 98 \
    > "$OUT/vitg14.json" 2> "$OUT/vitg14.err"
echo "vitg=$?" >> "$OUT/summary.txt"

timeout 600 python bench.py --arch vit_7b --patch-size 16 --grad-checkpointing \
    --global-size 512 --local-size 112 --batch-size 4 --local-crops 4 \
    --steps 3 --warmup 1 > "$OUT/vit7b_512.json" 2> "$OUT/vit7b_512.err"
echo "vit7b512=$?" >> "$OUT/summary.txt"

cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/$OUT/prof_g" -o g --output-format csv -- \
    python "$GRAFT_REPO_ROOT/bench.py" --arch vit_giant2 --patch-size 14 --local-size 98 \
    --batch-size 64 --steps 2 --warmup 1 > "$GRAFT_REPO_ROOT/$OUT/prof_g.log" 2>&1
echo "prof_g=$?" >> "$GRAFT_REPO_ROOT/$OUT/summary.txt"
cd "$GRAFT_REPO_ROOT"

cat "$OUT/summary.txt"
tail -1 "$OUT/vitg14.json"; grep peak "$OUT/vitg14.err"
tail -1 "$OUT/vit7b_512.json"; grep peak "$OUT/vit7b_512.err"; tail -2 "$OUT/vit7b_512.err"
