"""Profile harness: graphed decode kernel trace (see profiles/)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from tnn_amd import models
from tnn_amd.nn.layer import cast_compute_dtype
from tnn_amd.models.generate import generate_graphed

m = models.create_model("flash_gpt2_small")
cast_compute_dtype(m, torch.bfloat16)
m = m.to("cuda")
for rep in range(2):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = generate_graphed(m, list(range(16)), max_new_tokens=64, seq_len=512,
                           eot_token=None, device=torch.device("cuda"))
    torch.cuda.synchronize()
    print("decode 64 tok:", round(time.perf_counter() - t0, 3), "s")
