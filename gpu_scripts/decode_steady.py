"""Steady-state decode rate: second generate_graphed call after warm
capture infrastructure (the serving loop's regime)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tnn_amd import models
from tnn_amd.nn.layer import cast_compute_dtype
from tnn_amd.models.generate import generate_graphed

m = models.create_model("flash_gpt2_small")
cast_compute_dtype(m, torch.bfloat16)
m.to("cuda").eval()
for i in range(3):
    t0 = time.perf_counter()
    out = generate_graphed(m, list(range(16)), max_new_tokens=256,
                           seq_len=1024, eot_token=None)
    dt = time.perf_counter() - t0
    print(f"call {i}: {len(out)-16} tokens in {dt:.3f}s = "
          f"{(len(out)-16)/dt:.0f} tok/s")
