"""Flash attention autograd wrapper (hand-written CDNA4 kernels in
tnn_amd/csrc/attention.hip; replaces reference FlashAttentionBlock's
cuDNN-frontend SDPA graphs).

bf16 [B, H, S, D] with D in {64, 128} runs the fused kernels; anything
else (fp32 debug runs, odd head dims) falls back to a composed path.
"""

from __future__ import annotations

import torch

from .. import _C


class _FlashAttention(torch.autograd.Function):
    """q/k/v may be non-contiguous [B,H,S,D] views (transposed BSHD
    projections); the kernels read them through stride tuples and o and
    the grads come back as transposed views of BSHD buffers, so the whole
    attention round trip does zero layout copies (the round-2 profile
    showed ~8 activation-sized copies per layer per step without this)."""

    @staticmethod
    def forward(ctx, q, k, v, causal):
        ext = _C.ext()
        o, lse = ext.attn_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ext = _C.ext()
        dq, dk, dv = ext.attn_bwd(q, k, v, o, do, lse, ctx.causal)
        return dq, dk, dv, None


def _composed(q, k, v, causal):
    scale = q.shape[-1] ** -0.5
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        S = q.shape[-2]
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return (p @ v.float()).to(q.dtype)


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True) -> torch.Tensor:
    if (q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128)
            and q.is_cuda):
        return _FlashAttention.apply(q, k, v, causal)
    return _composed(q, k, v, causal)
