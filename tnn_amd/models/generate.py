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


@torch.no_grad()
def generate_graphed(model: Sequential, prompt_ids: List[int],
                     max_new_tokens: int = 50, seq_len: int = 1024,
                     eot_token: Optional[int] = 50256,
                     device: Optional[torch.device] = None,
                     use_graph: Optional[bool] = None) -> List[int]:
    """hipGraph-captured, fully GPU-resident greedy decode.

    The per-token step -- forward over fixed-capacity KV buffers, argmax,
    feeding the result back into the static input, recording it into a
    device id buffer, and advancing the device position counter -- is
    captured once and replayed per token with ZERO host round-trips;
    the generated ids are read back in one transfer at the end. EOT
    handling is post-hoc truncation (identical output to ``generate``).
    ``use_graph=False`` runs the identical tensor-driven step eagerly
    (CPU-testable path)."""
    from ..nn.blocks import _MHABase
    from ..nn.layers import PositionalEmbedding
    device = device or next(model.parameters()).device
    if use_graph is None:
        use_graph = device.type == "cuda"
    model.eval()
    attns = [m for m in model.modules() if isinstance(m, _MHABase)]
    poss = [m for m in model.modules() if isinstance(m, PositionalEmbedding)]
    pos_t = torch.zeros(1, dtype=torch.int64, device=device)
    for a in attns:
        a.enable_static_cache(seq_len, pos_t)
    try:
        ids = list(prompt_ids)
        plen = len(ids)
        total_new = min(max_new_tokens, seq_len - plen)
        if total_new <= 0:
            return ids
        x = torch.tensor([ids], dtype=torch.int64, device=device)
        logits = model(x)[0, -1]           # prefill (eager, fills caches)
        first = logits.argmax().reshape(1, 1)
        pos_t.fill_(plen)
        for p in poss:
            p._pos_tensor = pos_t
        static_tok = first.clone()
        ids_buf = torch.zeros(seq_len, dtype=torch.int64, device=device)
        ids_buf[plen] = first[0, 0]

        def step():
            out = model(static_tok)[:, -1]          # [1, V]
            nxt = out.argmax(-1, keepdim=True)      # [1, 1]
            pos_t.add_(1)
            ids_buf.index_copy_(0, pos_t, nxt[0])
            static_tok.copy_(nxt)

        n_replay = total_new - 1
        if use_graph and n_replay > 0:
            saved = (pos_t.clone(), static_tok.clone(), ids_buf.clone())
            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                for _ in range(2):      # warmup; stale cache rows stay masked
                    step()
            torch.cuda.current_stream().wait_stream(stream)
            pos_t.copy_(saved[0]); static_tok.copy_(saved[1])
            ids_buf.copy_(saved[2])
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                step()
            pos_t.copy_(saved[0]); static_tok.copy_(saved[1])
            ids_buf.copy_(saved[2])
            for _ in range(n_replay):
                graph.replay()
        else:
            for _ in range(n_replay):
                step()
        new_ids = ids_buf[plen:plen + total_new].tolist()
        if eot_token is not None and eot_token in new_ids:
            new_ids = new_ids[:new_ids.index(eot_token) + 1]
        return ids + new_ids
    finally:
        for a in attns:
            a.reset_cache()
        for p in poss:
            p._pos_tensor = None
