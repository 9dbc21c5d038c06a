#!/bin/bash
# PMC counter run for the conv kernels (own run, no trace domains mixed in).
set -x
REPO=$(cd "$(dirname "$0")/.." && pwd)
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

rocprofv3 --list-avail 2>/dev/null | grep -iE "BANK|MFMA|VALU_UTIL|LDS" | head -20 > "$OUT/pmc_avail.txt"

timeout 300 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_VALU_MFMA_BUSY_CYCLES \
    SQ_WAVE_CYCLES SQ_BUSY_CYCLES -d "$OUT/pmc1" -o pmc1 --output-format csv -- \
    python "$REPO/benchmarks/kernel_bench.py" conv > "$OUT/pmc1.log" 2>&1
tail -15 "$OUT/pmc1.log"
ls "$OUT/pmc1" || true
