#!/bin/bash
# PMC A/B of the 16-wide vs 32-wide NT GEMM cores (counters-only runs).
set -x
REPO=$(cd "$(dirname "$0")/.." && pwd)
OUT=$REPO/gpurun_out
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

cat > /tmp/one_gemm.py <<'PY'
import os, sys, torch
sys.path.insert(0, os.environ["REPO"])
from tnn_amd import _C
ext = _C.ext()
M, N, K = 4096, 3072, 768
a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
bn = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
for _ in range(30):
    c = ext.gemm_nt(a, bn)
torch.cuda.synchronize()
PY
for G in 1 0; do
  TNN_GEMM32=$G REPO=$REPO timeout 300 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT \
      SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_LDS \
      -d "$OUT/pmcg$G" -o pmcg$G --output-format csv -- \
      python /tmp/one_gemm.py > "$OUT/pmcg$G.log" 2>&1
done
ls "$OUT"/pmcg1 "$OUT"/pmcg0 || tail -5 "$OUT"/pmcg1.log
