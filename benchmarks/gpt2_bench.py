"""GPT-2 benchmarks (BASELINE config 4: GPT-2-small inference on 1x MI355X,
plus a training-step throughput measurement).

    python benchmarks/gpt2_bench.py [--model flash_gpt2_small] [--train]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnn_amd import models
from tnn_amd.nn import CrossEntropyLoss, AdamW
from tnn_amd.nn.layer import cast_compute_dtype
from tnn_amd.models.generate import generate, generate_cached, generate_graphed


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="flash_gpt2_small")
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--decode-tokens", type=int, default=32)
    p.add_argument("--train", action="store_true", default=True)
    p.add_argument("--graph", action="store_true",
                   help="also time the hipGraph-captured whole train step")
    args = p.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    model = models.create_model(args.model)
    if dev.type == "cuda":
        cast_compute_dtype(model, torch.bfloat16)
    model.to(dev)

    # ---- training-step throughput (tokens/s) ----
    if args.train and args.steps > 0:
        model.train()
        crit = CrossEntropyLoss()
        opt = AdamW(model.parameters(), lr=1e-4)
        x = torch.randint(0, 50257, (args.batch, args.seq_len), device=dev)
        y = torch.randint(0, 50257, (args.batch, args.seq_len), device=dev)

        def step():
            out = model(x)
            loss = crit(out, y)
            opt.zero_grad()
            loss.backward()
            opt.step()
            return loss

        # graph capture must see FRESH AccumulateGrad nodes (eager steps
        # first would pin them to the default stream and crash capture)
        if args.graph and dev.type == "cuda":
            from tnn_amd.utils.graphstep import GraphedTrainStep
            gstep = GraphedTrainStep(model, crit, opt, x, y)
            for _ in range(3):
                gstep()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                gloss = gstep()
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            toks = args.batch * args.seq_len * args.steps
            print(f"train[hipGraph]: {args.model} bs={args.batch} "
                  f"seq={args.seq_len}: {toks / dt:,.0f} tok/s "
                  f"({dt / args.steps * 1e3:.1f} ms/step, "
                  f"loss {gloss.item():.3f})")
            gstep.sync_step_count()
            opt._step_dev = None  # eager steps resume host step counting
            del gstep

        for _ in range(3):
            step()
        torch.cuda.synchronize() if dev.type == "cuda" else None
        t0 = time.perf_counter()
        for _ in range(args.steps):
            loss = step()
        torch.cuda.synchronize() if dev.type == "cuda" else None
        dt = time.perf_counter() - t0
        toks = args.batch * args.seq_len * args.steps
        print(f"train: {args.model} bs={args.batch} seq={args.seq_len}: "
              f"{toks / dt:,.0f} tok/s ({dt / args.steps * 1e3:.1f} ms/step, "
              f"loss {loss.item():.3f})")



    # ---- greedy decode (reference gpt2_inference loop) ----
    # note: at batch 1 and short sequences decode is kernel-launch-bound on
    # a 2.5PF GPU, so the full-recompute loop (one big batch of launches)
    # can beat the eager KV cache (many tiny launches); the hipGraph path
    # captures the whole per-token step into one graph launch.
    model.eval()
    for name, fn in [("recompute (reference parity)", generate),
                     ("kv-cache", generate_cached),
                     ("hipGraph", generate_graphed)]:
        t0 = time.perf_counter()
        out = fn(model, list(range(16)), max_new_tokens=args.decode_tokens,
                 seq_len=args.seq_len, device=dev, eot_token=None)
        dt = time.perf_counter() - t0
        n = len(out) - 16
        print(f"decode [{name}]: {n} tokens in {dt:.2f}s = {n / dt:.2f} tok/s")


if __name__ == "__main__":
    main()
