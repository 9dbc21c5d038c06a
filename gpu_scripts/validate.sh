#!/bin/bash
# GPU-box validation: numerics suite, benches, rocprof kernel stats.
# Run via: gpurun -- 'bash gpu_scripts/validate.sh'
set -x
REPO=$(cd "$(dirname "$0")/.." && pwd)
OUT=$REPO/gpurun_out
mkdir -p "$OUT"

cd "$REPO"
python -m pytest tests/test_gpu_numerics.py -q --timeout 900 2>&1 | tail -3

echo "=== BENCH wrn bf16 ==="
timeout 300 python bench.py --steps 20 --warmup 5 2>&1 | grep metric
echo "=== BENCH wrn fp32 ==="
timeout 300 python bench.py --steps 20 --warmup 5 --dtype fp32 2>&1 | grep metric
echo "=== BENCH resnet9 fp32 (BASELINE config 2) ==="
timeout 300 python bench.py --steps 20 --warmup 5 --model cifar10_resnet9 \
  --dtype fp32 2>&1 | grep metric

echo "=== PROFILE bf16 ==="
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats -d "$OUT/prof1" -o prof1 -- \
  python "$REPO/bench.py" --steps 5 --warmup 2 > "$OUT/prof1.log" 2>&1
cd "$REPO"
python gpu_scripts/prof_summary.py "$OUT/prof1/prof1_results.db" \
  "$OUT/prof1_summary.md" > /dev/null 2>&1
head -12 "$OUT/prof1_summary.md"
