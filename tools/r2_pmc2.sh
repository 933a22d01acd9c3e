#!/bin/bash
set -x
export TMPDIR=/tmp
cd /tmp
OUT="$GRAFT_REPO_ROOT/gpurun_out/r2_pmc2"
mkdir -p "$OUT"
timeout 420 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES GRBM_GUI_ACTIVE \
    -d "$OUT/pmc" -o mfma --output-format csv -- \
    python "$GRAFT_REPO_ROOT/bench.py" --steps 2 --warmup 1 > "$OUT/pmc.log" 2>&1
echo "pmc=$?" > "$OUT/summary.txt"
cat "$OUT/summary.txt"; ls "$OUT/pmc" | head -3
