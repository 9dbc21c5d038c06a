"""Primitive op layer (reference include/ops/ops.hpp:17-830 analog).

Every op has two implementations behind one functional API:

- **CPU**: a differentiable composition of PyTorch ops (the numerics
  oracle used by the test suite, reference §4's cross-backend parity
  strategy).
- **GPU (MI355X)**: hand-written CDNA4 HIP kernels from ``tnn_amd._hip``,
  wrapped in ``torch.autograd.Function`` so backward also runs our kernels.
  If the extension is missing on a GPU machine the op raises (no silent
  eager fallback).
"""

from .flash import flash_attention_qkv
from .functional import (
    conv2d_nhwc,
    batch_norm_act,
    linear,
    matmul,
    max_pool2d_nhwc,
    avg_pool2d_nhwc,
    dropout,
    activation,
    softmax_cross_entropy,
    layer_norm, layer_norm_res, linear_qkv,
    embedding,
    attention,
    sdpa_materialized,
    attention_decode,
    scaled_softmax,
    group_norm,
    pointwise_loss,
    add_act,
)

__all__ = [
    "conv2d_nhwc",
    "batch_norm_act",
    "linear",
    "matmul",
    "max_pool2d_nhwc",
    "avg_pool2d_nhwc",
    "dropout",
    "activation",
    "softmax_cross_entropy",
    "layer_norm",
    "embedding",
    "attention",
    "flash_attention_qkv",
    "sdpa_materialized",
    "attention_decode",
    "scaled_softmax",
    "group_norm",
    "pointwise_loss",
    "add_act",
]
