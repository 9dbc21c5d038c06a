"""Fluent model-construction DSL (reference include/nn/layer_builder.hpp:44-616).

Tracks the current batchless shape so layers get their input dims without
the user repeating them.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from .layer import Layer
from .layers import (Dense, Conv2D, BatchNorm, LayerNorm, GroupNorm, MaxPool2D,
                     AvgPool2D, Dropout, Activation, Flatten, Embedding,
                     PositionalEmbedding, ClassToken)
from .blocks import Sequential, ResidualBlock, GPTBlock


class LayerBuilder:
    def __init__(self, input_shape: Tuple[int, ...],
                 dtype: torch.dtype = torch.float32):
        """``input_shape`` is batchless: (H, W, C) for images, (S,) for tokens."""
        self.shape = tuple(input_shape)
        self._dtype = dtype
        self.layers: List[Layer] = []

    # -- infra ---------------------------------------------------------------
    def dtype(self, dt: torch.dtype) -> "LayerBuilder":
        self._dtype = dt
        return self

    def add(self, layer: Layer) -> "LayerBuilder":
        self.layers.append(layer)
        self.shape = layer.output_shape(self.shape)
        return self

    def build(self, name: str = "model") -> Sequential:
        return Sequential(self.layers, name=name)

    # -- primitive layers ----------------------------------------------------
    def conv2d(self, out_channels, kh=3, kw=3, sh=1, sw=1, ph=0, pw=0,
               bias=True, name="conv2d"):
        cin = self.shape[-1]
        return self.add(Conv2D(cin, out_channels, (kh, kw), (sh, sw), (ph, pw),
                               bias, name, self._dtype))

    def batchnorm(self, eps=1e-5, momentum=0.1, relu=False, name="batchnorm"):
        return self.add(BatchNorm(self.shape[-1], eps, momentum, relu, name,
                                  self._dtype))

    def layernorm(self, eps=1e-5, name="layernorm"):
        return self.add(LayerNorm(self.shape[-1], eps, name, self._dtype))

    def groupnorm(self, num_groups, eps=1e-5, name="groupnorm"):
        return self.add(GroupNorm(num_groups, self.shape[-1], eps, name, self._dtype))

    def maxpool2d(self, kh=2, kw=2, sh=None, sw=None, ph=0, pw=0, name="maxpool2d"):
        stride = (sh, sw) if sh is not None else None
        return self.add(MaxPool2D((kh, kw), stride, (ph, pw), name, self._dtype))

    def avgpool2d(self, kh=2, kw=2, sh=None, sw=None, ph=0, pw=0, name="avgpool2d"):
        stride = (sh, sw) if sh is not None else None
        return self.add(AvgPool2D((kh, kw), stride, (ph, pw), name, self._dtype))

    def dense(self, out_features, bias=True, name="dense"):
        return self.add(Dense(self.shape[-1], out_features, bias, name, self._dtype))

    def activation(self, kind="relu", name="activation"):
        return self.add(Activation(kind, name, self._dtype))

    def dropout(self, p=0.5, name="dropout"):
        return self.add(Dropout(p, name, self._dtype))

    def flatten(self, start_dim=1, end_dim=-1, name="flatten"):
        return self.add(Flatten(start_dim, end_dim, name, self._dtype))

    def embedding(self, vocab_size, dim, name="embedding"):
        return self.add(Embedding(vocab_size, dim, name, self._dtype))

    def positional_embedding(self, max_len, name="pos_embedding"):
        return self.add(PositionalEmbedding(max_len, self.shape[-1], name, self._dtype))

    def class_token(self, name="class_token"):
        return self.add(ClassToken(self.shape[-1], name, self._dtype))

    # -- composite blocks (reference layer_builder.hpp:339-436) --------------
    def basic_residual_block(self, in_channels, out_channels, stride=1,
                             name="basic_residual_block"):
        main = (LayerBuilder(self.shape, self._dtype)
                .conv2d(out_channels, 3, 3, stride, stride, 1, 1, False, f"{name}_conv1")
                .batchnorm(relu=True, name=f"{name}_bn0")
                .conv2d(out_channels, 3, 3, 1, 1, 1, 1, False, f"{name}_conv2")
                .batchnorm(relu=False, name=f"{name}_bn1")
                .build(f"{name}_main"))
        shortcut = None
        if stride != 1 or in_channels != out_channels:
            shortcut = (LayerBuilder(self.shape, self._dtype)
                        .conv2d(out_channels, 1, 1, stride, stride, 0, 0, False,
                                f"{name}_conv0")
                        .batchnorm(relu=False, name=f"{name}_bn_sc")
                        .build(f"{name}_shortcut"))
        return self.add(ResidualBlock(main, shortcut, "relu", name))

    def wide_residual_block(self, in_channels, out_channels, stride=1,
                            dropout_rate=0.0, name="wide_residual_block"):
        """Pre-activation WRN block: BN-ReLU → conv(s) → BN-ReLU → [dropout] →
        conv, linear join (reference layer_builder.hpp:367-405)."""
        b = (LayerBuilder(self.shape, self._dtype)
             .batchnorm(relu=True, name=f"{name}_bn1")
             .conv2d(out_channels, 3, 3, stride, stride, 1, 1, True, f"{name}_conv1")
             .batchnorm(relu=True, name=f"{name}_bn2"))
        if dropout_rate > 0:
            b.dropout(dropout_rate, f"{name}_drop")
        b.conv2d(out_channels, 3, 3, 1, 1, 1, 1, True, f"{name}_conv2")
        main = b.build(f"{name}_main")
        shortcut = None
        if stride != 1 or in_channels != out_channels:
            shortcut = (LayerBuilder(self.shape, self._dtype)
                        .conv2d(out_channels, 1, 1, stride, stride, 0, 0, False,
                                f"{name}_conv0")
                        .build(f"{name}_shortcut"))
        return self.add(ResidualBlock(main, shortcut, "linear", name))

    def bottleneck_residual_block(self, in_channels, mid_channels, out_channels,
                                  stride=1, name="bottleneck_residual_block"):
        main = (LayerBuilder(self.shape, self._dtype)
                .conv2d(mid_channels, 1, 1, 1, 1, 0, 0, False, f"{name}_conv1")
                .batchnorm(relu=True, name=f"{name}_bn0")
                .conv2d(mid_channels, 3, 3, stride, stride, 1, 1, False, f"{name}_conv2")
                .batchnorm(relu=True, name=f"{name}_bn1")
                .conv2d(out_channels, 1, 1, 1, 1, 0, 0, False, f"{name}_conv3")
                .batchnorm(relu=True, name=f"{name}_bn2")
                .build(f"{name}_main"))
        shortcut = None
        if stride != 1 or in_channels != out_channels:
            shortcut = (LayerBuilder(self.shape, self._dtype)
                        .conv2d(out_channels, 1, 1, stride, stride, 0, 0, False,
                                f"{name}_conv0")
                        .batchnorm(relu=False, name=f"{name}_bn_sc")
                        .build(f"{name}_shortcut"))
        return self.add(ResidualBlock(main, shortcut, "none", name))

    def gpt_block(self, num_heads, mlp_ratio=4, flash=True, dropout=0.0,
                  name="gpt_block"):
        return self.add(GPTBlock(self.shape[-1], num_heads, mlp_ratio, flash,
                                 dropout, name, self._dtype))
