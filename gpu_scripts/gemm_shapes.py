import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tnn_amd import _C
ext = _C.ext()
orig = ext.gemm
seen = {}
def wrap(a, b, *args, **kw):
    k = (a.shape[0], a.shape[1], b.shape[1])
    seen[k] = seen.get(k, 0) + 1
    return orig(a, b, *args, **kw)
ext.gemm = wrap
from tnn_amd import models
from tnn_amd.nn import CrossEntropyLoss, AdamW
from tnn_amd.nn.layer import cast_compute_dtype
m = models.create_model("flash_gpt2_small")
cast_compute_dtype(m, torch.bfloat16)
m.to("cuda").train()
crit, opt = CrossEntropyLoss(), AdamW(m.parameters(), lr=1e-4)
x = torch.randint(0, 50257, (8, 512), device="cuda")
y = torch.randint(0, 50257, (8, 512), device="cuda")
for _ in range(2):
    loss = crit(m(x), y); opt.zero_grad(); loss.backward(); opt.step()
torch.cuda.synchronize()
for k, v in sorted(seen.items(), key=lambda t: -t[1]):
    print(f"gemm M={k[0]} K={k[1]} N={k[2]} x{v}")
