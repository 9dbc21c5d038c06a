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


class _FlashQKV(torch.autograd.Function):
    """Flash attention straight off the merged projection buffer:
    qkv [B, S, 3*H*D] -> o [B, S, H*D]. q/k/v are head-slice VIEWS read
    through the kernels' stride tuples, and backward writes dq/dk/dv into
    ONE dqkv buffer (attn_bwd's dq_out/dk_out/dv_out) — without this the
    three slice consumers cost autograd 3 activation-size zero-fills +
    copies + 2 adds per layer per step."""

    @staticmethod
    def _views(t, H, D):
        HD = H * D
        return tuple(s.unflatten(-1, (H, D)).permute(0, 2, 1, 3)
                     for s in t.split(HD, dim=-1))

    @staticmethod
    def forward(ctx, qkv, H, causal):
        B, S, threeHD = qkv.shape
        D = threeHD // (3 * H)
        q, k, v = _FlashQKV._views(qkv, H, D)
        ext = _C.ext()
        o, lse = ext.attn_fwd(q, k, v, causal)
        ctx.save_for_backward(qkv, o, lse)
        ctx.H, ctx.D, ctx.causal = H, D, causal
        # o is a [B,H,S,D] view of a BSHD-contiguous buffer: this reshape
        # is free
        return o.transpose(1, 2).reshape(B, S, H * D)

    @staticmethod
    def backward(ctx, do):
        qkv, o, lse = ctx.saved_tensors
        H, D, causal = ctx.H, ctx.D, ctx.causal
        q, k, v = _FlashQKV._views(qkv, H, D)
        dqkv = torch.empty_like(qkv)
        dq_o, dk_o, dv_o = _FlashQKV._views(dqkv, H, D)
        dov = do.contiguous().unflatten(-1, (H, D)).permute(0, 2, 1, 3)
        ext = _C.ext()
        ext.attn_bwd(q, k, v, o, dov, lse, causal,
                     dq_out=dq_o, dk_out=dk_o, dv_out=dv_o)
        return dqkv, None, None


def flash_attention_qkv(qkv: torch.Tensor, num_heads: int,
                        causal: bool = True) -> torch.Tensor:
    """o [B,S,H*D] from a merged qkv [B,S,3*H*D] (see _FlashQKV)."""
    if qkv.is_cuda and qkv.dtype == torch.bfloat16:
        return _FlashQKV.apply(qkv, num_heads, causal)
    B, S, threeHD = qkv.shape
    D = threeHD // (3 * num_heads)
    q, k, v = (t.unflatten(-1, (num_heads, D)).permute(0, 2, 1, 3)
               for t in qkv.split(num_heads * D, dim=-1))
    o = _composed(q, k, v, causal).to(qkv.dtype)
    return o.transpose(1, 2).reshape(B, S, num_heads * D)


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
