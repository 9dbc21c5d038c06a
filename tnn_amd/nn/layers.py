"""Concrete layers (reference include/nn/layers_impl/ family, NHWC).

Every image layer takes/returns contiguous NHWC tensors. Conv weights are
``[KH, KW, Cin, Cout]`` — the implicit-GEMM B-operand layout consumed
directly by the CDNA4 MFMA conv kernels.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from .layer import Layer, register_layer


def _pair(v):
    return tuple(v) if isinstance(v, (tuple, list)) else (v, v)


@register_layer("dense")
class Dense(Layer):
    """Fully-connected layer (reference include/nn/layers_impl/dense_layer.hpp).

    Weight stored ``[in_features, out_features]`` row-major — the GEMM
    B operand, no transpose needed on the hot path.
    """

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 name: str = "dense", dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.in_features, self.out_features = in_features, out_features
        k = 1.0 / math.sqrt(in_features)
        self.weight = nn.Parameter(torch.empty(in_features, out_features,
                                               dtype=dtype).uniform_(-k, k))
        self.bias = nn.Parameter(torch.empty(out_features, dtype=dtype)
                                 .uniform_(-k, k)) if bias else None

    def forward(self, x):
        return ops.linear(x.to(self.io_dtype), self.weight, self.bias)

    def output_shape(self, in_shape):
        return (*in_shape[:-1], self.out_features)

    def flops_per_item(self, in_shape):
        lead = math.prod(in_shape[:-1]) if len(in_shape) > 1 else 1
        return 2 * lead * self.in_features * self.out_features

    def extra_config(self):
        return {"in_features": self.in_features, "out_features": self.out_features,
                "bias": self.bias is not None}


@register_layer("conv2d")
class Conv2D(Layer):
    """NHWC convolution (reference include/nn/layers_impl/conv2d_layer.hpp)."""

    def __init__(self, in_channels: int, out_channels: int,
                 kernel=(3, 3), stride=(1, 1), padding=(0, 0), bias: bool = True,
                 name: str = "conv2d", dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.in_channels, self.out_channels = in_channels, out_channels
        self.kernel, self.stride, self.padding = _pair(kernel), _pair(stride), _pair(padding)
        kh, kw = self.kernel
        fan_in = kh * kw * in_channels
        std = math.sqrt(2.0 / fan_in)  # He init for ReLU nets
        self.weight = nn.Parameter(
            torch.randn(kh, kw, in_channels, out_channels, dtype=dtype) * std)
        self.bias = nn.Parameter(torch.zeros(out_channels, dtype=dtype)) if bias else None

    def forward(self, x):
        return ops.conv2d_nhwc(x.to(self.io_dtype), self.weight, self.bias,
                               self.stride, self.padding)

    def output_shape(self, in_shape):
        h, w, _ = in_shape
        kh, kw = self.kernel
        oh = (h + 2 * self.padding[0] - kh) // self.stride[0] + 1
        ow = (w + 2 * self.padding[1] - kw) // self.stride[1] + 1
        return (oh, ow, self.out_channels)

    def flops_per_item(self, in_shape):
        oh, ow, co = self.output_shape(in_shape)
        return 2 * oh * ow * co * self.kernel[0] * self.kernel[1] * self.in_channels

    def extra_config(self):
        return {"in_channels": self.in_channels, "out_channels": self.out_channels,
                "kernel": list(self.kernel), "stride": list(self.stride),
                "padding": list(self.padding), "bias": self.bias is not None}


@register_layer("batchnorm")
class BatchNorm(Layer):
    """BatchNorm over the NHWC channel dim with optional fused ReLU.

    gamma/beta/running stats are always fp32 regardless of io dtype
    (reference src/nn/layers_impl/batchnorm_layer.cpp:140-148); the fused
    ReLU flag mirrors the reference's BN+ReLU graphs
    (cudnn_batchnorm_ops.cu:159).
    """

    def __init__(self, num_features: int, eps: float = 1e-5, momentum: float = 0.1,
                 relu: bool = False, name: str = "batchnorm",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.num_features, self.eps, self.momentum, self.relu = num_features, eps, momentum, relu
        self.gamma = nn.Parameter(torch.ones(num_features, dtype=torch.float32))
        self.beta = nn.Parameter(torch.zeros(num_features, dtype=torch.float32))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def forward(self, x):
        return ops.batch_norm_act(x, self.gamma, self.beta,
                                  self.running_mean, self.running_var,
                                  self.training, self.momentum, self.eps, self.relu)

    def forward_res(self, x):
        """(bn(x), residual passthrough of x): pre-activation residual
        blocks route their shortcut through the second output so the
        junction's grad join fuses into bn_bwd (see _BatchNormActRes)."""
        if x.is_cuda and self.training:
            return ops.batch_norm_act(x, self.gamma, self.beta,
                                      self.running_mean, self.running_var,
                                      True, self.momentum, self.eps,
                                      self.relu, passthrough=True)
        return self.forward(x), x

    def flops_per_item(self, in_shape):
        return 8 * math.prod(in_shape)

    def extra_config(self):
        return {"num_features": self.num_features, "eps": self.eps,
                "momentum": self.momentum, "relu": self.relu}


@register_layer("layernorm")
class LayerNorm(Layer):
    def __init__(self, dim: int, eps: float = 1e-5, name: str = "layernorm",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.dim, self.eps = dim, eps
        self.gamma = nn.Parameter(torch.ones(dim, dtype=torch.float32))
        self.beta = nn.Parameter(torch.zeros(dim, dtype=torch.float32))

    def forward(self, x):
        return ops.layer_norm(x, self.gamma, self.beta, self.eps)

    def flops_per_item(self, in_shape):
        return 8 * math.prod(in_shape)

    def extra_config(self):
        return {"dim": self.dim, "eps": self.eps}


@register_layer("groupnorm")
class GroupNorm(Layer):
    """GroupNorm over NHWC (reference groupnorm_ops.cu:46)."""

    def __init__(self, num_groups: int, num_channels: int, eps: float = 1e-5,
                 name: str = "groupnorm", dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.num_groups, self.num_channels, self.eps = num_groups, num_channels, eps
        self.gamma = nn.Parameter(torch.ones(num_channels, dtype=torch.float32))
        self.beta = nn.Parameter(torch.zeros(num_channels, dtype=torch.float32))

    def forward(self, x):
        return ops.group_norm(x, self.gamma, self.beta, self.num_groups,
                              self.eps)

    def extra_config(self):
        return {"num_groups": self.num_groups, "num_channels": self.num_channels,
                "eps": self.eps}


@register_layer("maxpool2d")
class MaxPool2D(Layer):
    def __init__(self, kernel=(2, 2), stride=None, padding=(0, 0),
                 name: str = "maxpool2d", dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.kernel = _pair(kernel)
        self.stride = _pair(stride) if stride is not None else self.kernel
        self.padding = _pair(padding)

    def forward(self, x):
        return ops.max_pool2d_nhwc(x, self.kernel, self.stride, self.padding)

    def output_shape(self, in_shape):
        h, w, c = in_shape
        oh = (h + 2 * self.padding[0] - self.kernel[0]) // self.stride[0] + 1
        ow = (w + 2 * self.padding[1] - self.kernel[1]) // self.stride[1] + 1
        return (oh, ow, c)

    def flops_per_item(self, in_shape):
        return math.prod(in_shape)

    def extra_config(self):
        return {"kernel": list(self.kernel), "stride": list(self.stride),
                "padding": list(self.padding)}


@register_layer("avgpool2d")
class AvgPool2D(MaxPool2D):
    def __init__(self, kernel=(2, 2), stride=None, padding=(0, 0),
                 name: str = "avgpool2d", dtype: torch.dtype = torch.float32):
        super().__init__(kernel, stride, padding, name, dtype)

    def forward(self, x):
        return ops.avg_pool2d_nhwc(x, self.kernel, self.stride, self.padding)


@register_layer("dropout")
class Dropout(Layer):
    def __init__(self, p: float = 0.5, name: str = "dropout",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.p = p

    def forward(self, x):
        return ops.dropout(x, self.p, self.training)

    def extra_config(self):
        return {"p": self.p}


@register_layer("activation")
class Activation(Layer):
    def __init__(self, kind: str = "relu", name: str = "activation",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.kind = kind

    def forward(self, x):
        return ops.activation(x, self.kind)

    def extra_config(self):
        return {"kind": self.kind}


@register_layer("flatten")
class Flatten(Layer):
    def __init__(self, start_dim: int = 1, end_dim: int = -1, name: str = "flatten",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.start_dim, self.end_dim = start_dim, end_dim

    def forward(self, x):
        return torch.flatten(x, self.start_dim, self.end_dim)

    def output_shape(self, in_shape):
        # in_shape is batchless; layer dims are full-tensor dims (dim 0 = batch)
        nd = len(in_shape) + 1
        start = self.start_dim % nd
        end = self.end_dim % nd
        s, e = start - 1, end - 1  # batchless indices
        return (*in_shape[:s], math.prod(in_shape[s:e + 1]), *in_shape[e + 1:])

    def extra_config(self):
        return {"start_dim": self.start_dim, "end_dim": self.end_dim}


@register_layer("identity")
class Identity(Layer):
    def forward(self, x):
        return x


@register_layer("transpose")
class Transpose(Layer):
    def __init__(self, dim0: int, dim1: int, name: str = "transpose",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.dim0, self.dim1 = dim0, dim1

    def forward(self, x):
        return x.transpose(self.dim0, self.dim1).contiguous()

    def extra_config(self):
        return {"dim0": self.dim0, "dim1": self.dim1}


@register_layer("slice")
class Slice(Layer):
    """Slice along a dim (reference slice_ops.cu)."""

    def __init__(self, dim: int, start: int, end: Optional[int] = None,
                 name: str = "slice", dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.dim, self.start, self.end = dim, start, end

    def forward(self, x):
        return x.narrow(self.dim, self.start,
                        (self.end if self.end is not None else x.shape[self.dim]) - self.start)

    def output_shape(self, in_shape):
        if self.dim == 0 or self.end is None:
            return tuple(in_shape)
        s = list(in_shape)
        s[self.dim - 1] = self.end - self.start
        return tuple(s)

    def extra_config(self):
        return {"dim": self.dim, "start": self.start, "end": self.end}


@register_layer("embedding")
class Embedding(Layer):
    """Token embedding gather (reference embedding_ops.cu:17)."""

    def __init__(self, vocab_size: int, dim: int, name: str = "embedding",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.vocab_size, self.dim = vocab_size, dim
        self.weight = nn.Parameter(torch.randn(vocab_size, dim, dtype=dtype) * 0.02)

    def forward(self, ids):
        return ops.embedding(ids, self.weight)

    def output_shape(self, in_shape):
        return (*in_shape, self.dim)

    def extra_config(self):
        return {"vocab_size": self.vocab_size, "dim": self.dim}


@register_layer("positional_embedding")
class PositionalEmbedding(Layer):
    """Learned positional embedding added to [B, S, D] input."""

    def __init__(self, max_len: int, dim: int, name: str = "pos_embedding",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.max_len, self.dim = max_len, dim
        self.weight = nn.Parameter(torch.randn(max_len, dim, dtype=dtype) * 0.02)

    _pos_offset = 0      # set by cached decode (models/generate.py)
    _pos_tensor = None   # device int64 [1]: graph-capturable position

    def forward(self, x):
        s = x.shape[-2]
        if self._pos_tensor is not None and s == 1:
            return x + self.weight.index_select(0, self._pos_tensor)
        off = self._pos_offset
        return x + self.weight[off:off + s]

    def extra_config(self):
        return {"max_len": self.max_len, "dim": self.dim}


@register_layer("class_token")
class ClassToken(Layer):
    """Prepend a learned CLS token (reference class_token_ops.cu:18)."""

    def __init__(self, dim: int, name: str = "class_token",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.dim = dim
        self.token = nn.Parameter(torch.zeros(1, 1, dim, dtype=dtype))

    def forward(self, x):
        b = x.shape[0]
        return torch.cat([self.token.expand(b, 1, self.dim).to(x.dtype), x], dim=1)

    def output_shape(self, in_shape):
        s, d = in_shape
        return (s + 1, d)

    def extra_config(self):
        return {"dim": self.dim}


@register_layer("nary")
class NAry(Layer):
    """Elementwise join of multiple inputs (reference n_ary_ops.cu:32-163:
    add/sub/mul/div joins). Takes a list/tuple of tensors."""

    OPS = {"add": torch.add, "sub": torch.sub, "mul": torch.mul,
           "div": torch.div}

    def __init__(self, op: str = "add", name: str = "nary",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.op = op

    def forward(self, *xs):
        if len(xs) == 1 and isinstance(xs[0], (list, tuple)):
            xs = xs[0]
        fn = self.OPS[self.op]
        y = xs[0]
        for x in xs[1:]:
            y = fn(y, x)
        return y

    def extra_config(self):
        return {"op": self.op}


@register_layer("mbroadcast")
class MBroadcast(Layer):
    """Fan-out one input to N consumers; backward sums the incoming grads
    (reference include/nn/layers_impl/mbroadcast_layer.hpp:12). With
    autograd the fan-out is just returning the tensor N times — grads sum
    automatically."""

    def __init__(self, n: int = 2, name: str = "mbroadcast",
                 dtype: torch.dtype = torch.float32):
        super().__init__(name, dtype)
        self.n = n

    def forward(self, x):
        return tuple(x for _ in range(self.n))

    def extra_config(self):
        return {"n": self.n}
