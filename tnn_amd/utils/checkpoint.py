"""Checkpoint archive format (reference include/nn/graph.hpp:119-183,
include/tensor/tensor.hpp:585-606).

Same structural *idea* as the reference — a self-describing JSON header
(the model's full config) followed by raw per-tensor records in
registration order — but not byte-compatible: the reference streams JSON
unprefixed and writes ndims as u64, ours length-prefixes the header and
uses ``{u32 dtype, u32 ndims, u64 shape[], bytes}`` records with its own
dtype codes. We additionally store tensor names in the header (the
reference relies purely on registration order) and, unlike the reference,
can also checkpoint optimizer state (the reference does not — SURVEY §5).

Layout:
    [u64 header_len][header JSON utf-8][record 0][record 1]...
"""

from __future__ import annotations

import json
import struct
from typing import Any, BinaryIO, Dict, List, Optional, Tuple

import numpy as np
import torch

_DTYPE_CODES = {
    torch.uint8: 0,
    torch.int32: 1,
    torch.int64: 2,
    torch.float16: 3,
    torch.bfloat16: 4,
    torch.float32: 5,
    torch.float64: 6,
    torch.bool: 7,
}
_CODE_DTYPES = {v: k for k, v in _DTYPE_CODES.items()}


def _write_tensor(f: BinaryIO, t: torch.Tensor):
    t = t.detach().contiguous().cpu()
    f.write(struct.pack("<II", _DTYPE_CODES[t.dtype], t.dim()))
    f.write(struct.pack(f"<{t.dim()}Q", *t.shape) if t.dim() else b"")
    if t.dtype == torch.bfloat16:
        data = t.view(torch.uint16).numpy()
    else:
        data = t.numpy()
    f.write(data.tobytes())


def _read_tensor(f: BinaryIO) -> torch.Tensor:
    code, ndims = struct.unpack("<II", f.read(8))
    shape = struct.unpack(f"<{ndims}Q", f.read(8 * ndims)) if ndims else ()
    dtype = _CODE_DTYPES[code]
    numel = 1
    for s in shape:
        numel *= s
    nbytes = numel * torch.empty(0, dtype=dtype).element_size()
    raw = f.read(nbytes)
    if dtype == torch.bfloat16:
        arr = np.frombuffer(raw, dtype=np.uint16).copy()
        t = torch.from_numpy(arr).view(torch.bfloat16)
    else:
        t = torch.from_numpy(np.frombuffer(raw, dtype=_np_dtype(dtype)).copy())
    return t.reshape(shape)


def _np_dtype(dt: torch.dtype):
    return {torch.uint8: np.uint8, torch.int32: np.int32, torch.int64: np.int64,
            torch.float16: np.float16, torch.float32: np.float32,
            torch.float64: np.float64, torch.bool: np.bool_}[dt]


def _model_tensors(model: torch.nn.Module) -> List[Tuple[str, torch.Tensor]]:
    out = list(model.named_parameters())
    out += [(n, b) for n, b in model.named_buffers()]
    return out


def save_model(model, path: str, extra_header: Optional[Dict[str, Any]] = None):
    tensors = _model_tensors(model)
    header = {
        "format": "tnn_amd.checkpoint.v1",
        "config": model.get_config() if hasattr(model, "get_config") else None,
        "tensors": [n for n, _ in tensors],
    }
    if extra_header:
        header.update(extra_header)
    blob = json.dumps(header).encode()
    with open(path, "wb") as f:
        f.write(struct.pack("<Q", len(blob)))
        f.write(blob)
        for _, t in tensors:
            _write_tensor(f, t)


def _read_header(f: BinaryIO) -> Dict[str, Any]:
    (hlen,) = struct.unpack("<Q", f.read(8))
    return json.loads(f.read(hlen).decode())


def load_model(path: str, model: Optional[torch.nn.Module] = None):
    """Load weights into ``model``, or rebuild the model from the stored
    config when ``model`` is None (reference Graph::load_state)."""
    with open(path, "rb") as f:
        header = _read_header(f)
        if model is None:
            from ..nn.layer import layer_from_config
            model = layer_from_config(header["config"])
        tensors = dict(_model_tensors(model))
        with torch.no_grad():
            for name in header["tensors"]:
                t = _read_tensor(f)
                dst = tensors[name]
                dst.copy_(t.to(dst.dtype))
    return model


def save_checkpoint(model, optimizer, path: str,
                    extra: Optional[Dict[str, Any]] = None):
    """Model + optimizer state (exceeds the reference, which drops Adam m/v)."""
    opt_tensors = optimizer.state_tensors() if optimizer is not None else []
    tensors = _model_tensors(model) + [(f"opt/{n}", t) for n, t in opt_tensors]
    header = {
        "format": "tnn_amd.checkpoint.v1",
        "config": model.get_config() if hasattr(model, "get_config") else None,
        "optimizer": optimizer.get_config() if optimizer is not None else None,
        "step_count": getattr(optimizer, "step_count", 0),
        "tensors": [n for n, _ in tensors],
        **(extra or {}),
    }
    blob = json.dumps(header).encode()
    with open(path, "wb") as f:
        f.write(struct.pack("<Q", len(blob)))
        f.write(blob)
        for _, t in tensors:
            _write_tensor(f, t)


def load_checkpoint(path: str, model, optimizer=None) -> Dict[str, Any]:
    """Restore model weights + optimizer state. A freshly constructed
    optimizer (the normal resume flow) has empty lazy state; its slots are
    materialized from the record names before copying, so Adam m/v and fp32
    master weights survive resume. Records that cannot be placed raise —
    silently dropping optimizer moments would give oversized momentum-free
    post-resume steps."""
    unplaced = []
    with open(path, "rb") as f:
        header = _read_header(f)
        named = dict(_model_tensors(model))
        opt_named = {}
        if optimizer is not None:
            opt_named = {f"opt/{n}": t for n, t in optimizer.state_tensors()}
        with torch.no_grad():
            for name in header["tensors"]:
                t = _read_tensor(f)
                dst = named.get(name)
                if dst is None:
                    dst = opt_named.get(name)
                if dst is None and name.startswith("opt/") and optimizer is not None \
                        and hasattr(optimizer, "materialize_state_slot"):
                    dst = optimizer.materialize_state_slot(name[len("opt/"):])
                if dst is not None:
                    dst.copy_(t.to(dst.dtype))
                elif not (name.startswith("opt/") and optimizer is None):
                    # opt/ records are skipped intentionally when no
                    # optimizer was passed (model-only load)
                    unplaced.append(name)
    if unplaced:
        raise RuntimeError(
            f"checkpoint {path!r}: {len(unplaced)} tensor record(s) could not "
            f"be placed: {unplaced[:8]}{'...' if len(unplaced) > 8 else ''}")
    if optimizer is not None:
        optimizer.step_count = header.get("step_count", 0)
    return header
