"""NN library (reference include/nn/ analog, re-designed for PyTorch-ROCm).

Layers are ``torch.nn.Module`` subclasses with TNN-style define-by-config
serialization (``get_config``/``from_config``), shape inference for the
pipeline partitioner, and NHWC image layout. Autograd replaces the
reference's hand-rolled per-layer backward + per-micro-batch caches: a
pipeline stage keeps one autograd graph per in-flight micro-batch instead
(reference include/nn/layer.hpp:113 mb_id caches).
"""

from .layer import Layer, register_layer, layer_from_config, LAYER_REGISTRY
from .layers import (
    Dense, Conv2D, BatchNorm, LayerNorm, GroupNorm,
    MaxPool2D, AvgPool2D, Dropout, Activation, Flatten,
    Embedding, PositionalEmbedding, ClassToken, Identity, Transpose, Slice,
    NAry, MBroadcast,
)
from .blocks import Sequential, ResidualBlock, MSequential, AttentionBlock, FlashAttentionBlock, GPTBlock
from .builder import LayerBuilder
from .losses import Loss, CrossEntropyLoss, MSELoss, MAELoss, HuberLoss, loss_from_config
from .optim import Optimizer, SGD, Adam, AdamW, optimizer_from_config
from .schedulers import scheduler_from_config
from . import schedulers
from .accuracy import accuracy
from .train import TrainingConfig, train_model, validate_model
from .graph import Graph, GraphBuilder, GraphExecutor

__all__ = [
    "Layer", "register_layer", "layer_from_config", "LAYER_REGISTRY",
    "Dense", "Conv2D", "BatchNorm", "LayerNorm", "GroupNorm",
    "MaxPool2D", "AvgPool2D", "Dropout", "Activation", "Flatten",
    "Embedding", "PositionalEmbedding", "ClassToken", "Identity", "Transpose", "Slice",
    "NAry", "MBroadcast",
    "Sequential", "ResidualBlock", "MSequential", "AttentionBlock",
    "FlashAttentionBlock", "GPTBlock", "LayerBuilder",
    "Loss", "CrossEntropyLoss", "MSELoss", "MAELoss", "HuberLoss", "loss_from_config",
    "Optimizer", "SGD", "Adam", "AdamW", "optimizer_from_config",
    "scheduler_from_config", "schedulers", "accuracy",
    "TrainingConfig", "train_model", "validate_model",
    "Graph", "GraphBuilder", "GraphExecutor",
]
