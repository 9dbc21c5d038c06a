"""Kernel micro-benchmarks (reference benchmarks/{gemm,conv2d,dense,
batchnorm,...}_benchmark.cpp analog).

    python benchmarks/kernel_bench.py [op ...]   # default: all

Prints one line per case: op, shape, ms, TFLOP/s (or GB/s for memory ops).
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tnn_amd import _C  # noqa: E402

ext = _C.ext() if torch.cuda.is_available() else None
DEV = "cuda"


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def report(name, shape, secs, flops=None, bytes_=None):
    extra = ""
    if flops:
        extra = f"{flops / secs / 1e12:8.2f} TF/s"
    if bytes_:
        extra += f"{bytes_ / secs / 1e12:8.2f} TB/s"
    print(f"{name:18s} {shape:42s} {secs * 1e3:8.3f} ms {extra}")


def bench_gemm(dtype=torch.bfloat16):
    for M, N, K in [(4096, 4096, 4096), (65536, 128, 1152), (16384, 512, 4608),
                    (8192, 768, 768), (256, 100, 512)]:
        a = torch.randn(M, K, dtype=dtype, device=DEV)
        b = torch.randn(K, N, dtype=dtype, device=DEV)
        secs = timeit(lambda: ext.gemm(a, b, None, 0))
        report("gemm_nn", f"{M}x{N}x{K} {dtype}", secs, 2.0 * M * N * K)


def bench_conv(dtype=torch.bfloat16):
    cases = [
        ("g1 conv 128->128", 256, 32, 32, 128, 128, 3, 1, 1),
        ("g2 conv 256->256", 256, 16, 16, 256, 256, 3, 1, 1),
        ("g3 conv 512->512", 256, 8, 8, 512, 512, 3, 1, 1),
        ("stride2 128->256", 256, 32, 32, 128, 256, 3, 2, 1),
    ]
    for name, N, H, W, Ci, Co, KS, S, P in cases:
        x = torch.randn(N, H, W, Ci, dtype=dtype, device=DEV)
        w = torch.randn(KS, KS, Ci, Co, dtype=dtype, device=DEV) * 0.05
        OH = (H + 2 * P - KS) // S + 1
        dy = torch.randn(N, OH, OH, Co, dtype=dtype, device=DEV)
        fl = 2.0 * N * OH * OH * Co * KS * KS * Ci
        report(f"conv_fwd", f"{name} mb{N}", timeit(lambda: ext.conv2d_fwd(
            x, w, None, S, S, P, P, False)), fl)
        report(f"conv_dgrad", f"{name} mb{N}", timeit(lambda: ext.conv2d_dgrad(
            dy, w, H, W, S, S, P, P)), fl)
        report(f"conv_wgrad", f"{name} mb{N}", timeit(lambda: ext.conv2d_wgrad(
            x, dy, KS, KS, S, S, P, P, False)), fl)


def bench_bn(dtype=torch.bfloat16):
    for C in [128, 512]:
        rows = 256 * 32 * 32 if C == 128 else 256 * 8 * 8
        x = torch.randn(rows // (8 * 8), 8, 8, C, dtype=dtype, device=DEV)
        g = torch.ones(C, device=DEV)
        b = torch.zeros(C, device=DEV)
        nbytes = x.numel() * x.element_size()
        secs = timeit(lambda: ext.bn_fwd_train(x, g, b, None, None, 0.1, 1e-5, True, 0.0, 0, None))
        report("bn_fwd_train", f"rows={rows} C={C} {dtype}", secs,
               bytes_=3 * nbytes)


def bench_colsum(dtype=torch.bfloat16):
    x = torch.randn(256 * 32 * 32, 128, dtype=dtype, device=DEV)
    secs = timeit(lambda: ext.colsum(x))
    report("colsum", f"{tuple(x.shape)} {dtype}", secs,
           bytes_=x.numel() * x.element_size())


def bench_ce(dtype=torch.bfloat16):
    logits = torch.randn(8192, 50257, dtype=dtype, device=DEV)
    t = torch.randint(0, 50257, (8192,), device=DEV)
    secs = timeit(lambda: ext.ce_fwd(logits, t))
    report("ce_fwd", "8192x50257", secs,
           bytes_=logits.numel() * logits.element_size())


def bench_attn():
    for B, H, S, D, causal in [(8, 12, 512, 64, True), (8, 12, 1024, 64, True),
                               (2, 16, 2048, 128, False)]:
        q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=DEV)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
        fl = 4.0 * B * H * S * S * D * (0.5 if causal else 1.0)
        secs = timeit(lambda: ext.attn_fwd(q, k, v, causal))
        report("attn_fwd", f"B{B} H{H} S{S} D{D} causal={causal}", secs, fl)
        o, lse = ext.attn_fwd(q, k, v, causal)
        do = torch.randn_like(q)
        secs = timeit(lambda: ext.attn_bwd(q, k, v, o, do, lse, causal),
                      iters=20)
        report("attn_bwd", f"B{B} H{H} S{S} D{D} causal={causal}", secs,
               2.5 * fl)


def bench_bmm():
    # materialized-scores attention shapes (per-head batched QK^T)
    for BH, S, D in [(96, 512, 64), (32, 2048, 128)]:
        a = torch.randn(BH, S, D, dtype=torch.bfloat16, device=DEV)
        b = torch.randn(BH, S, D, dtype=torch.bfloat16, device=DEV)
        fl = 2.0 * BH * S * S * D
        secs = timeit(lambda: ext.bmm(a, b.transpose(-1, -2)))
        report("bmm_nt", f"BH{BH} S{S} D{D} (QK^T)", secs, fl)


def bench_softmax():
    from tnn_amd import ops
    for BH, S in [(96, 512), (32, 2048)]:
        x = torch.randn(BH, S, S, dtype=torch.bfloat16, device=DEV)
        by = 2 * x.numel() * 2
        secs = timeit(lambda: ext.smax_fwd(x, S, 0, 0.125, True))
        report("smax_fwd", f"BH{BH} S{S} causal", secs, bytes_=by)


def bench_ln():
    for R, C in [(4096, 768), (8192, 1024)]:
        x = torch.randn(R, C, dtype=torch.bfloat16, device=DEV)
        g = torch.randn(C, device=DEV)
        b = torch.randn(C, device=DEV)
        by = 2 * x.numel() * 2
        secs = timeit(lambda: ext.ln_fwd(x, g, b, 1e-5))
        report("ln_fwd", f"{R}x{C}", secs, bytes_=by)


def bench_gn():
    for N, HW, C, G in [(256, 64, 256, 32)]:
        x = torch.randn(N, HW, C, dtype=torch.bfloat16, device=DEV)
        g = torch.randn(C, device=DEV)
        b = torch.randn(C, device=DEV)
        by = 2 * x.numel() * 2
        secs = timeit(lambda: ext.gn_fwd(x, g, b, G, 1e-5))
        report("gn_fwd", f"N{N} HW{HW} C{C} G{G}", secs, bytes_=by)


def bench_decode():
    from tnn_amd import ops
    BH, cap, D = 12, 1024, 64
    q = torch.randn(BH, D, dtype=torch.bfloat16, device=DEV)
    kc = torch.randn(BH, cap, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    secs = timeit(lambda: ops.attention_decode(q, kc, vc, None, cap,
                                               D ** -0.5))
    report("attn_decode", f"BH{BH} cap{cap} D{D}", secs,
           bytes_=2 * BH * cap * D * 2)


ALL = {"gemm": bench_gemm, "conv": bench_conv, "bn": bench_bn,
       "colsum": bench_colsum, "ce": bench_ce, "attn": bench_attn,
       "bmm": bench_bmm, "softmax": bench_softmax, "ln": bench_ln,
       "gn": bench_gn, "decode": bench_decode}

if __name__ == "__main__":
    which = sys.argv[1:] or list(ALL)
    for name in which:
        ALL[name]()
