"""A/B timing of the NT GEMM paths on transformer shapes (TNN_GEMM32 env)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tnn_amd import _C
ext = _C.ext()

def t(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

shapes = [(4096, 768, 768), (4096, 3072, 768), (4096, 768, 3072),
          (4096, 2304, 768), (4096, 50257, 768)]
for M, N, K in shapes:
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(K, N, dtype=torch.bfloat16, device="cuda")
    bn = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    us_g = t(lambda: ext.gemm(a, b, None, 0))
    us_nt = t(lambda: ext.gemm_nt(a, bn))
    tf = 2 * M * N * K / 1e12
    print(f"M{M} N{N} K{K}: gemm {us_g:8.1f}us {tf/us_g*1e6:6.0f}TF | "
          f"gemm_nt {us_nt:8.1f}us {tf/us_nt*1e6:6.0f}TF")
