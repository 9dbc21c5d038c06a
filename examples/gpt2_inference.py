#!/usr/bin/env python3
"""GPT-2 greedy decode demo (reference examples/gpt2_inference.cpp:19-127).

    python examples/gpt2_inference.py --model flash_gpt2_small \
        [--vocab vocab.bin] [--snapshot path] --prompt "Hello"

Without a vocab/snapshot this runs with random weights and raw token ids —
the compute path (embedding -> N gpt blocks -> ln_f -> head, full-sequence
recompute per token, matching the reference's no-KV-cache loop).
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnn_amd import models
from tnn_amd.data import Tokenizer
from tnn_amd.models.generate import generate, generate_cached
from tnn_amd.nn.layer import cast_compute_dtype
from tnn_amd.utils.checkpoint import load_model


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="flash_gpt2_small")
    p.add_argument("--snapshot", default=None)
    p.add_argument("--vocab", default=None)
    p.add_argument("--prompt", default="Hello world")
    p.add_argument("--tokens", type=int, default=50)
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--bf16", action="store_true")
    p.add_argument("--kv-cache", action="store_true",
                   help="cached decode (the reference recomputes)")
    args = p.parse_args()

    model = (load_model(args.snapshot) if args.snapshot
             else models.create_model(args.model))
    if args.bf16:
        cast_compute_dtype(model, torch.bfloat16)
    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    model.to(dev)

    tok = Tokenizer().load(args.vocab) if args.vocab else None
    prompt_ids = tok.encode(args.prompt) if tok else list(range(10))

    gen = generate_cached if args.kv_cache else generate
    t0 = time.perf_counter()
    out = gen(model, prompt_ids, max_new_tokens=args.tokens,
              seq_len=args.seq_len, device=dev,
              eot_token=50256 if tok else None)
    dt = time.perf_counter() - t0
    n_new = len(out) - len(prompt_ids)
    mode = "kv-cache" if args.kv_cache else "full-sequence recompute"
    print(f"{n_new} tokens in {dt:.2f}s ({n_new / dt:.2f} tok/s, {mode})")
    print(tok.decode(out) if tok else out)


if __name__ == "__main__":
    main()
