#!/bin/bash
# GPU-box validation: numerics suite, bf16+fp32 bench, rocprof kernel stats.
# Run via: gpurun -- 'bash gpu_scripts/validate.sh'
set -x
REPO=$(cd "$(dirname "$0")/.." && pwd)
OUT=$REPO/gpurun_out
mkdir -p "$OUT"

cd "$REPO"
python -m pytest tests/test_gpu_numerics.py -q --timeout 900 2>&1 | tail -4

echo "=== BENCH bf16 ==="
timeout 300 python bench.py --steps 20 --warmup 5 2>&1 | tail -2
echo "=== BENCH fp32 ==="
timeout 300 python bench.py --steps 20 --warmup 5 --dtype fp32 2>&1 | tail -1

echo "=== PROFILE bf16 ==="
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d "$OUT/prof1" -o prof1 -- \
  python "$REPO/bench.py" --steps 5 --warmup 2 > "$OUT/prof1.log" 2>&1
echo "--- top kernels ---"
find "$OUT/prof1" -name "*stats*" | head -5
for f in $(find "$OUT/prof1" -name "*kernel_stats*"); do head -25 "$f"; done
tail -5 "$OUT/prof1.log"
