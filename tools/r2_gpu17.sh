#!/bin/bash
# Round-2 GPU call #17: model-zoo breadth on hardware — every ViT size ctor
# steps on device, ConvNeXt forward works, and the 768px high-res shape
# (N=2305, hd-128) runs a real train step.
set -x
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu17
mkdir -p "$OUT"

for arch in vit_base vit_so400m vit_huge2; do
  timeout 300 python bench.py --arch $arch --steps 3 --warmup 1 --batch-size 32 \
      > "$OUT/bench_$arch.json" 2> "$OUT/bench_$arch.err"
  echo "$arch=$?" >> "$OUT/summary.txt"
done

# 768px high-res train step (ViT-L/16 backbone at 2304 patch tokens, the
# high-res-adapt shape; hd-64) and a vit_7b 768px forward would OOM the
# teacher+student at batch>1 — step vit_large @768 batch 4
timeout 300 python bench.py --arch vit_large --global-size 768 --local-size 192 \
    --batch-size 4 --local-crops 2 --steps 3 --warmup 1 --grad-checkpointing \
    > "$OUT/bench_hires.json" 2> "$OUT/bench_hires.err"
echo "hires=$?" >> "$OUT/summary.txt"

# ConvNeXt forward on device
timeout 240 python - > "$OUT/convnext.log" 2>&1 <<'PY'
import torch
from dinov3_amd.models.convnext import convnext_tiny
m = convnext_tiny().cuda().bfloat16().eval()
x = torch.randn(4, 3, 224, 224, device="cuda").bfloat16()
out = m(x)
feats = out["x_norm_clstoken"] if isinstance(out, dict) else out
print("convnext ok", {k: tuple(v.shape) for k, v in out.items()} if isinstance(out, dict) else feats.shape)
PY
echo "convnext=$?" >> "$OUT/summary.txt"

cat "$OUT/summary.txt"
for f in "$OUT"/bench_*.json; do echo "$f"; tail -1 "$f"; done
grep peak "$OUT"/*.err 2>/dev/null
tail -2 "$OUT/convnext.log"
