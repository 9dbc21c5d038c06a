"""Model zoo (reference src/nn/example_models.cpp:21-559, registered :529).

Every model is a serializable Sequential built through LayerBuilder; the
registry + ``load_or_create_model`` mirror reference
include/nn/example_models.hpp:49.
"""

from __future__ import annotations

import os
from typing import Callable, Dict

import torch

from ..nn.builder import LayerBuilder
from ..nn.blocks import Sequential

_REGISTRY: Dict[str, Callable[..., Sequential]] = {}


def register_model(name: str):
    def deco(fn):
        _REGISTRY[name] = fn
        return fn
    return deco


def model_names():
    return sorted(_REGISTRY)


def create_model(name: str, dtype: torch.dtype = torch.float32) -> Sequential:
    return _REGISTRY[name](dtype=dtype)


def load_or_create_model(name: str, snapshot_dir: str = "model_snapshots",
                         dtype: torch.dtype = torch.float32) -> Sequential:
    """reference include/nn/example_models.hpp:49."""
    path = os.path.join(snapshot_dir, name)
    if os.path.exists(path):
        from ..utils.checkpoint import load_model
        return load_model(path)
    return create_model(name, dtype)


# ---------------------------------------------------------------------------
# CNN family (reference example_models.cpp:21-283)
# ---------------------------------------------------------------------------


@register_model("mnist_cnn")
def mnist_cnn(dtype=torch.float32):
    return (LayerBuilder((28, 28, 1), dtype)
            .conv2d(8, 5, 5, 1, 1, 0, 0, False, "conv1")
            .batchnorm(relu=True, name="bn1")
            .maxpool2d(3, 3, 3, 3, 0, 0, "pool1")
            .conv2d(16, 1, 1, 1, 1, 0, 0, False, "conv2_1x1")
            .batchnorm(relu=True, name="bn2_1x1")
            .activation("relu", "relu2")
            .conv2d(48, 5, 5, 1, 1, 0, 0, False, "conv3")
            .batchnorm(relu=True, name="bn3")
            .maxpool2d(2, 2, 2, 2, 0, 0, "pool2")
            .flatten(1, -1, "flatten")
            .dense(10, False, "output")
            .build("mnist_cnn"))


@register_model("cifar10_vgg")
def cifar10_vgg(dtype=torch.float32):
    b = LayerBuilder((32, 32, 3), dtype)
    plan = [(64, True), (64, True), ("pool", None), (128, True), (128, True),
            ("pool", None), (256, True), (256, "relu"), (256, True),
            ("pool", None), (512, True), (512, True), (512, True), ("pool", None)]
    ci = 0
    for spec, bn in plan:
        if spec == "pool":
            b.maxpool2d(2, 2, 2, 2, 0, 0, f"pool{ci}")
        else:
            b.conv2d(spec, 3, 3, 1, 1, 1, 1, False, f"conv{ci}")
            if bn is True:
                b.batchnorm(relu=True, name=f"bn{ci}")
            elif bn == "relu":
                b.activation("relu", f"relu{ci}")
            ci += 1
    return (b.flatten(1, -1, "flatten")
            .dense(512, True, "fc0").activation("relu", "relu_fc")
            .dense(10, True, "fc1")
            .build("cifar10_vgg"))


@register_model("cifar10_resnet9")
def cifar10_resnet9(dtype=torch.float32):
    return (LayerBuilder((32, 32, 3), dtype)
            .conv2d(64, 3, 3, 1, 1, 1, 1, False, "conv1")
            .batchnorm(relu=True, name="bn1")
            .conv2d(128, 3, 3, 1, 1, 1, 1, False, "conv2")
            .batchnorm(relu=True, name="bn2")
            .maxpool2d(2, 2, 2, 2, 0, 0, "pool1")
            .basic_residual_block(128, 128, 1, "res_block1")
            .basic_residual_block(128, 128, 1, "res_block2")
            .conv2d(256, 3, 3, 1, 1, 1, 1, False, "conv3")
            .batchnorm(relu=True, name="bn3")
            .maxpool2d(2, 2, 2, 2, 0, 0, "pool2")
            .basic_residual_block(256, 256, 1, "res_block3")
            .basic_residual_block(256, 256, 1, "res_block4")
            .conv2d(512, 3, 3, 1, 1, 1, 1, False, "conv4")
            .batchnorm(relu=True, name="bn4")
            .maxpool2d(2, 2, 2, 2, 0, 0, "pool3")
            .basic_residual_block(512, 512, 1, "res_block5")
            .avgpool2d(4, 4, 1, 1, 0, 0, "avgpool")
            .flatten(1, -1, "flatten")
            .dense(10, True, "output")
            .build("cifar10_resnet9"))


def _resnet18(input_hw: int, num_classes: int, final_pool: int, dtype):
    b = (LayerBuilder((input_hw, input_hw, 3), dtype)
         .conv2d(32, 3, 3, 1, 1, 1, 1, False, "conv1")
         .batchnorm(relu=True, name="bn1")
         .maxpool2d(2, 2, 2, 2, 0, 0, "maxpool")
         .basic_residual_block(32, 64, 1, "layer1_block1")
         .basic_residual_block(64, 64, 1, "layer1_block2")
         .basic_residual_block(64, 128, 2, "layer2_block1")
         .basic_residual_block(128, 128, 1, "layer2_block2")
         .basic_residual_block(128, 256, 2, "layer3_block1")
         .basic_residual_block(256, 256, 1, "layer3_block2")
         .basic_residual_block(256, 512, 2, "layer4_block1")
         .basic_residual_block(512, 512, 1, "layer4_block2")
         .avgpool2d(final_pool, final_pool, 1, 1, 0, 0, "avgpool")
         .flatten(1, -1, "flatten")
         .dense(num_classes, True, "fc"))
    return b


@register_model("cifar100_resnet18")
def cifar100_resnet18(dtype=torch.float32):
    return _resnet18(32, 100, 2, dtype).build("cifar100_resnet18")


@register_model("tiny_imagenet_resnet18")
def tiny_imagenet_resnet18(dtype=torch.float32):
    return _resnet18(64, 200, 4, dtype).build("tiny_imagenet_resnet18")


def _wrn16_8(input_hw: int, num_classes: int, dtype):
    c1, c2, c3 = 128, 256, 512
    dropout_rate = 0.3
    return (LayerBuilder((input_hw, input_hw, 3), dtype)
            .conv2d(16, 3, 3, 1, 1, 1, 1, True, "conv1")
            .wide_residual_block(16, c1, 1, dropout_rate, "group1_block1")
            .wide_residual_block(c1, c1, 1, dropout_rate, "group1_block2")
            .wide_residual_block(c1, c2, 2, dropout_rate, "group2_block1")
            .wide_residual_block(c2, c2, 1, dropout_rate, "group2_block2")
            .wide_residual_block(c2, c3, 2, dropout_rate, "group3_block1")
            .wide_residual_block(c3, c3, 1, dropout_rate, "group3_block2")
            .batchnorm(relu=True, name="bn_final")
            .avgpool2d(input_hw // 4, input_hw // 4, 1, 1, 0, 0, "avgpool")
            .flatten(1, -1, "flatten")
            .dense(num_classes, True, "fc"))


@register_model("cifar100_wrn16_8")
def cifar100_wrn16_8(dtype=torch.float32):
    """The headline pipeline-benchmark model
    (reference example_models.cpp:130, sample_logs/cifar100_wrn16_8)."""
    return _wrn16_8(32, 100, dtype).build("cifar100_wrn16_8")


@register_model("tiny_imagenet_wrn16_8")
def tiny_imagenet_wrn16_8(dtype=torch.float32):
    return _wrn16_8(64, 200, dtype).build("tiny_imagenet_wrn16_8")


def _resnet50(input_hw, num_classes, stem, final_pool, dtype):
    b = LayerBuilder((input_hw, input_hw, 3), dtype)
    if stem == "imagenet":
        b.conv2d(64, 7, 7, 2, 2, 3, 3, True, "conv1")
    else:
        b.conv2d(64, 3, 3, 1, 1, 1, 1, True, "conv1")
    b.batchnorm(relu=True, name="bn1").maxpool2d(3, 3, 2, 2, 1, 1, "maxpool")
    layout = [(64, 256, 3, 1), (128, 512, 4, 2), (256, 1024, 6, 2),
              (512, 2048, 3, 2)]
    cin = 64
    for li, (mid, out, blocks, stride) in enumerate(layout, start=1):
        for bi in range(blocks):
            s = stride if bi == 0 else 1
            b.bottleneck_residual_block(cin, mid, out, s,
                                        f"layer{li}_block{bi + 1}")
            cin = out
    return (b.avgpool2d(final_pool, final_pool, 1, 1, 0, 0, "avgpool")
            .flatten(1, -1, "flatten")
            .dense(num_classes, True, "fc"))


@register_model("tiny_imagenet_resnet50")
def tiny_imagenet_resnet50(dtype=torch.float32):
    return _resnet50(64, 200, "tiny", 4, dtype).build("tiny_imagenet_resnet50")


@register_model("imagenet_resnet50")
def imagenet_resnet50(dtype=torch.float32):
    return _resnet50(224, 1000, "imagenet", 7, dtype).build("imagenet_resnet50")


# ---------------------------------------------------------------------------
# ViT family (reference example_models.cpp:286-383)
# ---------------------------------------------------------------------------


def _vit(flash: bool, dtype):
    from ..nn.blocks import ResidualBlock, AttentionBlock, FlashAttentionBlock
    from ..nn.layers import LayerNorm, Dropout
    patch, embed_dim, heads, mlp_ratio, depth, classes = 4, 256, 4, 4, 4, 200
    seq_len = (64 // patch) ** 2 + 1
    b = (LayerBuilder((64, 64, 3), dtype)
         .conv2d(embed_dim, patch, patch, patch, patch, 0, 0, True, "patch_embed")
         .flatten(1, 2, "flatten_patches"))  # [N,16,16,D] -> [N,256,D]
    b.class_token("class_token")
    b.positional_embedding(seq_len, "pos_embed")
    b.dropout(0.1, "drop_in")
    attn_cls = FlashAttentionBlock if flash else AttentionBlock
    for i in range(depth):
        attn_main = Sequential([
            LayerNorm(embed_dim, name=f"enc{i}_ln_attn", dtype=dtype),
            attn_cls(embed_dim, heads, causal=False, name=f"enc{i}_attn",
                     dtype=dtype),
            Dropout(0.1, name=f"enc{i}_attn_drop", dtype=dtype),
        ], name=f"enc{i}_attn_main")
        b.add(ResidualBlock(attn_main, None, "linear", f"encoder_{i}_attn"))
        mlp = (LayerBuilder((seq_len, embed_dim), dtype)
               .layernorm(name=f"enc{i}_ln_mlp")
               .dense(embed_dim * mlp_ratio, False, f"enc{i}_fc1")
               .activation("gelu", f"enc{i}_gelu")
               .dropout(0.1, f"enc{i}_mlp_drop1")
               .dense(embed_dim, False, f"enc{i}_fc2")
               .dropout(0.1, f"enc{i}_mlp_drop2")
               .build(f"enc{i}_mlp_main"))
        b.add(ResidualBlock(mlp, None, "linear", f"encoder_{i}_mlp"))
    from ..nn.layers import Slice
    b.layernorm(name="ln_final")
    b.add(Slice(1, 0, 1, "extract_cls"))
    return (b.flatten(1, -1, "flatten_cls")
            .dense(classes, True, "head"))


@register_model("tiny_imagenet_vit")
def tiny_imagenet_vit(dtype=torch.float32):
    return _vit(False, dtype).build("tiny_imagenet_vit")


@register_model("tiny_imagenet_flash_vit")
def tiny_imagenet_flash_vit(dtype=torch.float32):
    return _vit(True, dtype).build("tiny_imagenet_flash_vit")


# ---------------------------------------------------------------------------
# GPT-2 family (reference example_models.cpp:384-528); seq 1024, vocab 50257
# ---------------------------------------------------------------------------


def _gpt2(embed_dim, heads, layers, flash, name, dtype, seq_len=1024,
          vocab=50257, dropout=0.1):
    b = (LayerBuilder((seq_len,), dtype)
         .embedding(vocab, embed_dim, "token_embed")
         .positional_embedding(seq_len, "pos_embed")
         .dropout(dropout, "drop_in"))
    for i in range(layers):
        b.gpt_block(heads, 4, flash, dropout, f"block{i}")
    return (b.layernorm(name="ln_f")
            .dense(vocab, True, "head")
            .build(name))


for _name, _dims in [("gpt2_small", (768, 12, 12)), ("gpt2_medium", (1024, 16, 24)),
                     ("gpt2_large", (1280, 20, 36))]:
    def _mk(dims=_dims, name=_name, flash=False):
        def f(dtype=torch.float32):
            return _gpt2(*dims, flash, name if not flash else f"flash_{name}", dtype)
        return f
    _REGISTRY[_name] = _mk()
    _REGISTRY[f"flash_{_name}"] = _mk(flash=True)
