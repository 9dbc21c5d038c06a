"""Layer base class + config registry.

Mirrors the reference's ``Layer`` contract (reference include/nn/layer.hpp:44):
named layers, dtype policy, config round-trip via a factory
(reference include/nn/layers.hpp:95 ``LayerFactory``), and a shape-inference
hook used by the pipeline partitioner (the reference infers output shapes in
``Worker::process_message``, include/distributed/worker.hpp:145).

Unlike the reference there is no hand-written backward: layers are
``torch.nn.Module``s and autograd provides the backward pass; micro-batch
activation caches live in the autograd graphs a pipeline stage keeps per
in-flight micro-batch.
"""

from __future__ import annotations

from typing import Any, Dict, Tuple

import torch
import torch.nn as nn

LAYER_REGISTRY: Dict[str, type] = {}

_DTYPES = {
    "float32": torch.float32,
    "bfloat16": torch.bfloat16,
    "float16": torch.float16,
    "float64": torch.float64,
}
_DTYPE_NAMES = {v: k for k, v in _DTYPES.items()}
_DTYPES.update({"fp32": torch.float32, "bf16": torch.bfloat16,
                "fp16": torch.float16})  # CLI-style aliases


def dtype_from_name(name: str) -> torch.dtype:
    return _DTYPES[name]


def dtype_name(dt: torch.dtype) -> str:
    return _DTYPE_NAMES[dt]


def register_layer(type_name: str):
    """Class decorator registering a layer type for config-driven construction."""

    def deco(cls):
        cls._type = type_name
        LAYER_REGISTRY[type_name] = cls
        return cls

    return deco


def layer_from_config(cfg: Dict[str, Any]) -> "Layer":
    cfg = dict(cfg)
    type_name = cfg.pop("type")
    cls = LAYER_REGISTRY[type_name]
    return cls.from_config(cfg)


def cast_compute_dtype(model: nn.Module, dtype: torch.dtype) -> nn.Module:
    """Cast compute params (conv/dense weights...) to ``dtype`` while norm
    affine params (named gamma/beta) and buffers stay fp32 — the mixed
    precision policy (reference keeps BN params fp32,
    batchnorm_layer.cpp:140-148)."""
    for mod in model.modules():
        for name, p in mod.named_parameters(recurse=False):
            if p.dtype == torch.float32 and name not in ("gamma", "beta"):
                p.data = p.data.to(dtype)
        if hasattr(mod, "io_dtype"):
            mod.io_dtype = dtype
    return model


class Layer(nn.Module):
    """Base layer: named, dtype-aware, config round-trippable."""

    _type = "layer"

    def __init__(self, name: str = "", dtype: torch.dtype = torch.float32):
        super().__init__()
        self.name = name or self._type
        self.io_dtype = dtype

    # -- config round-trip ---------------------------------------------------
    def extra_config(self) -> Dict[str, Any]:
        """Layer-specific constructor arguments (overridden by subclasses)."""
        return {}

    def get_config(self) -> Dict[str, Any]:
        cfg = {"type": self._type, "name": self.name,
               "dtype": dtype_name(self.io_dtype)}
        cfg.update(self.extra_config())
        return cfg

    @classmethod
    def from_config(cls, cfg: Dict[str, Any]) -> "Layer":
        cfg = dict(cfg)
        cfg.pop("type", None)
        if "dtype" in cfg and isinstance(cfg["dtype"], str):
            cfg["dtype"] = dtype_from_name(cfg["dtype"])
        return cls(**cfg)

    # -- shape inference (batchless shapes, e.g. (H, W, C) or (S,)) ----------
    def output_shape(self, in_shape: Tuple[int, ...]) -> Tuple[int, ...]:
        return tuple(in_shape)

    # -- bookkeeping ----------------------------------------------------------
    def param_count(self) -> int:
        return sum(p.numel() for p in self.parameters())

    def flops_per_item(self, in_shape: Tuple[int, ...]) -> int:
        """Approximate forward FLOPs for one batch item; partitioner weight."""
        return 0

    def __repr__(self):  # keep reprs short; nn.Module default is noisy
        return f"{type(self).__name__}(name={self.name!r})"
