"""Autoregressive generation (reference examples/gpt2_inference.cpp:19-127).

The reference recomputes the full sequence per token (:71-122, explicitly
no KV cache). ``generate`` reproduces that; ``generate_cached`` adds the
KV-cache fast path the reference lacks, exploiting that every layer in the
GPT zoo is causal: past activations are position-invariant, so we only
recompute the suffix window when the prompt overflows.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from ..nn.blocks import Sequential


@torch.no_grad()
def generate(model: Sequential, prompt_ids: List[int], max_new_tokens: int = 50,
             seq_len: int = 1024, eot_token: Optional[int] = 50256,
             device: Optional[torch.device] = None,
             greedy: bool = True, temperature: float = 1.0) -> List[int]:
    """Greedy/sampled decode with full-sequence recompute per token."""
    device = device or next(model.parameters()).device
    model.eval()
    ids = list(prompt_ids)
    for _ in range(max_new_tokens):
        window = ids[-seq_len:]
        x = torch.tensor([window], dtype=torch.int64, device=device)
        logits = model(x)[0, len(window) - 1]
        if greedy:
            nxt = int(logits.argmax().item())
        else:
            probs = torch.softmax(logits.float() / temperature, dim=-1)
            nxt = int(torch.multinomial(probs, 1).item())
        ids.append(nxt)
        if eot_token is not None and nxt == eot_token:
            break
    return ids


@torch.no_grad()
def generate_cached(model: Sequential, prompt_ids: List[int],
                    max_new_tokens: int = 50, seq_len: int = 1024,
                    eot_token: Optional[int] = 50256,
                    device: Optional[torch.device] = None,
                    greedy: bool = True, temperature: float = 1.0) -> List[int]:
    """KV-cached decode: prefill once, then one token per step."""
    from ..nn.blocks import _MHABase
    from ..nn.layers import PositionalEmbedding
    device = device or next(model.parameters()).device
    model.eval()
    attns = [m for m in model.modules() if isinstance(m, _MHABase)]
    poss = [m for m in model.modules() if isinstance(m, PositionalEmbedding)]
    for a in attns:
        a.enable_cache(max_len=seq_len)
    try:
        ids = list(prompt_ids)
        x = torch.tensor([ids], dtype=torch.int64, device=device)
        logits = model(x)[0, -1]
        for t in range(max_new_tokens):
            if greedy:
                nxt = int(logits.argmax().item())
            else:
                probs = torch.softmax(logits.float() / temperature, dim=-1)
                nxt = int(torch.multinomial(probs, 1).item())
            ids.append(nxt)
            if eot_token is not None and nxt == eot_token:
                break
            if len(ids) >= seq_len:
                break
            for p in poss:
                p._pos_offset = len(ids) - 1
            step = torch.tensor([[nxt]], dtype=torch.int64, device=device)
            logits = model(step)[0, -1]
        return ids
    finally:
        for a in attns:
            a.reset_cache()
        for p in poss:
            p._pos_offset = 0
