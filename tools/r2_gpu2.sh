#!/bin/bash
# Round-2 GPU call #2: P=14 patch-embed validation, full suite regression,
# hd-128 high-res FMHA numbers, ViT-g/14 occupancy rehearsal (config #5),
# and a PMC pass for the GEMM roofline evidence.
set -x
export TMPDIR=/tmp
cd /tmp && cd "$GRAFT_REPO_ROOT" || cd /root/repo
OUT=gpurun_out/r2_gpu2
mkdir -p "$OUT"

# 1. full GPU suite (includes the new P=14/P=8 patch-embed cases)
timeout 420 python -m pytest tests -m gpu -q > "$OUT/pytest.log" 2>&1
echo "gpu_suite=$?" >> "$OUT/summary.txt"

# 2. kernel micro-benchmarks incl. hd-128 @ N=2305
timeout 240 python tools/bench_kernels.py --iters 30 > "$OUT/bench_kernels.log" 2>&1
echo "kernel_bench=$?" >> "$OUT/summary.txt"

# 3. ViT-g/14 + activation checkpointing occupancy rehearsal (config #5)
timeout 420 python bench.py --arch vit_giant2 --patch-size 14 --grad-checkpointing \
    --steps 4 --warmup 2 --batch-size 64 > "$OUT/bench_vitg14.json" 2> "$OUT/bench_vitg14.err"
echo "vitg14=$?" >> "$OUT/summary.txt"

# 4. PMC pass: MFMA busy / wave cycles / GUI_ACTIVE per kernel on the
#    headline step (counters only — never combined with trace domains)
cd /tmp
timeout 420 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES GRBM_GUI_ACTIVE \
    -d "$GRAFT_REPO_ROOT/$OUT/pmc" -o mfma --output-format csv -- \
    python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 \
    > "$GRAFT_REPO_ROOT/$OUT/pmc.log" 2>&1
echo "pmc=$?" >> "$GRAFT_REPO_ROOT/$OUT/summary.txt"
cd "$GRAFT_REPO_ROOT"

cat "$OUT/summary.txt"
tail -5 "$OUT/pytest.log"
cat "$OUT/bench_vitg14.json" "$OUT/bench_vitg14.err" 2>/dev/null | tail -5
ls -la "$OUT/pmc" 2>/dev/null | head
