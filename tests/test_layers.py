"""Layer-level tests: forward shapes match shape inference, config
round-trips rebuild identical structures (reference unit-test pattern,
unit_tests/*_layer_test.cpp)."""

import pytest
import torch

from tnn_amd import nn as tnn
from tnn_amd.nn.layer import layer_from_config


def roundtrip(layer):
    cfg = layer.get_config()
    rebuilt = layer_from_config(cfg)
    assert rebuilt.get_config() == cfg
    return rebuilt


CASES = [
    (lambda: tnn.Dense(16, 8, True, "d"), (4, 16)),
    (lambda: tnn.Conv2D(3, 8, (3, 3), (1, 1), (1, 1), True, "c"), (2, 8, 8, 3)),
    (lambda: tnn.Conv2D(3, 8, (3, 3), (2, 2), (1, 1), False, "c"), (2, 9, 9, 3)),
    (lambda: tnn.Conv2D(4, 8, (1, 1), (2, 2), (0, 0), False, "c"), (2, 8, 8, 4)),
    (lambda: tnn.BatchNorm(6, relu=True), (2, 5, 5, 6)),
    (lambda: tnn.LayerNorm(12), (2, 7, 12)),
    (lambda: tnn.GroupNorm(2, 8), (2, 4, 4, 8)),
    (lambda: tnn.MaxPool2D((2, 2)), (2, 8, 8, 3)),
    (lambda: tnn.AvgPool2D((3, 3), (2, 2), (1, 1)), (2, 9, 9, 3)),
    (lambda: tnn.Dropout(0.5), (2, 10)),
    (lambda: tnn.Activation("gelu"), (2, 10)),
    (lambda: tnn.Flatten(), (2, 4, 4, 3)),
    (lambda: tnn.Flatten(1, 2), (2, 4, 4, 3)),
    (lambda: tnn.Embedding(50, 16), None),
    (lambda: tnn.PositionalEmbedding(32, 16), (2, 10, 16)),
    (lambda: tnn.ClassToken(16), (2, 10, 16)),
    (lambda: tnn.Slice(1, 0, 1), (2, 10, 16)),
    (lambda: tnn.Transpose(1, 2), (2, 5, 7)),
    (lambda: tnn.Identity(), (2, 3)),
]


@pytest.mark.parametrize("make,shape", CASES,
                         ids=[c[0]().name if hasattr(c[0](), "name") else str(i)
                              for i, c in enumerate(CASES)])
def test_layer_forward_and_roundtrip(make, shape):
    layer = make()
    rebuilt = roundtrip(layer)
    if shape is None:  # embedding takes int ids
        x = torch.randint(0, 50, (2, 10))
    else:
        x = torch.randn(*shape)
    layer.eval()
    y = layer(x)
    inferred = layer.output_shape(tuple(x.shape[1:]))
    if not isinstance(layer, (tnn.Transpose,)):
        assert tuple(y.shape[1:]) == tuple(inferred), type(layer).__name__
    # rebuilt layer accepts the same input
    rebuilt.eval()
    y2 = rebuilt(x)
    assert y2.shape == y.shape


def test_batchnorm_running_stats_and_relu():
    bn = tnn.BatchNorm(4, relu=True)
    bn.train()
    x = torch.randn(8, 3, 3, 4) * 3 + 1
    y = bn(x)
    assert (y >= 0).all()
    assert not torch.allclose(bn.running_mean, torch.zeros(4))
    # eval mode uses running stats
    bn.eval()
    y2 = bn(x)
    assert y2.shape == x.shape


def test_batchnorm_matches_torch():
    bn = tnn.BatchNorm(5)
    ref = torch.nn.BatchNorm2d(5, eps=bn.eps, momentum=bn.momentum)
    x = torch.randn(6, 4, 4, 5)
    bn.train(), ref.train()
    y = bn(x)
    yref = ref(x.permute(0, 3, 1, 2)).permute(0, 2, 3, 1)
    assert torch.allclose(y, yref, atol=1e-5)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-5)


def test_conv_matches_torch():
    conv = tnn.Conv2D(3, 8, (3, 3), (1, 1), (1, 1), True)
    x = torch.randn(2, 8, 8, 3)
    y = conv(x)
    ref = torch.nn.functional.conv2d(
        x.permute(0, 3, 1, 2), conv.weight.permute(3, 2, 0, 1), conv.bias,
        padding=1).permute(0, 2, 3, 1)
    assert torch.allclose(y, ref, atol=1e-5)


def test_sequential_slice_and_config():
    seq = (tnn.LayerBuilder((8, 8, 3))
           .conv2d(4, 3, 3, 1, 1, 1, 1, True, "c1")
           .batchnorm(relu=True, name="b1")
           .maxpool2d(2, 2)
           .flatten()
           .dense(10)
           .build("m"))
    assert seq.output_shape((8, 8, 3)) == (10,)
    left, right = seq.slice(0, 2), seq.slice(2, len(seq))
    x = torch.randn(2, 8, 8, 3)
    seq.eval(), left.eval(), right.eval()
    assert torch.allclose(right(left(x)), seq(x), atol=1e-6)
    cfg = seq.get_config()
    rebuilt = layer_from_config(cfg)
    assert rebuilt.get_config() == cfg


def test_residual_block_shapes():
    b = (tnn.LayerBuilder((8, 8, 16))
         .basic_residual_block(16, 32, 2, "rb")
         .build())
    x = torch.randn(2, 8, 8, 16)
    b.eval()
    assert b(x).shape == (2, 4, 4, 32)
    assert b.output_shape((8, 8, 16)) == (4, 4, 32)


def test_nary_and_mbroadcast():
    from tnn_amd.nn import NAry, MBroadcast
    a, b = torch.randn(3, 4), torch.randn(3, 4)
    assert torch.allclose(NAry("add")(a, b), a + b)
    assert torch.allclose(NAry("mul")([a, b]), a * b)
    x = torch.randn(3, 4, requires_grad=True)
    outs = MBroadcast(3)(x)
    assert len(outs) == 3
    sum(o.sum() for o in outs).backward()
    assert torch.allclose(x.grad, torch.full_like(x, 3.0))


def test_msequential_exec_order_preserves_join_semantics():
    """Memory-aware scheduling may run branches out of declaration order,
    but non-commutative joins must still combine outputs in declaration
    order (reference msequential.cpp compute_execution_order)."""
    import torch
    from tnn_amd.nn.blocks import MSequential
    from tnn_amd.nn.layer import Layer

    class Scale(Layer):
        def __init__(self, f):
            super().__init__(f"scale{f}", torch.float32)
            self.f = f

        def forward(self, x):
            return x * self.f

        def output_shape(self, in_shape):
            return tuple(in_shape)

    m = MSequential([Scale(3.0), Scale(2.0)], join="sub")
    x = torch.ones(2, 4)
    y = m(x)
    assert torch.allclose(y, torch.full((2, 4), 1.0))  # 3x - 2x, not 2x - 3x
    # force a reversed execution order: declaration-order join must hold
    m2 = MSequential([Scale(3.0), Scale(2.0)], join="sub")
    m2._exec_order = [1, 0]
    assert torch.allclose(m2(x), torch.full((2, 4), 1.0))


@pytest.mark.gpu
def test_msequential_memory_order_measured_on_gpu():
    import torch
    from tnn_amd.nn.blocks import MSequential
    from tnn_amd.nn.layers import Dense
    wide = Dense(64, 512, name="wide")    # big transient, big retained
    slim = Dense(64, 8, name="slim")
    m = MSequential([slim, wide], join="concat").to("cuda")
    x = torch.randn(16, 64, device="cuda")
    y = m(x)
    assert y.shape == (16, 520)
    assert sorted(m._exec_order) == [0, 1]  # an order was measured
    # joins still concat in declaration order
    ref = torch.cat([m.branches[0](x), m.branches[1](x)], dim=-1)
    assert torch.allclose(y, ref)


def test_linear_qkv_cpu_fallback():
    import torch
    from tnn_amd import ops
    torch.manual_seed(5)
    x = torch.randn(2, 8, 16)
    ws = [torch.randn(16, 16) for _ in range(3)]
    bs = [torch.randn(16) for _ in range(3)]
    y = ops.linear_qkv(x, *ws, *bs)
    ref = torch.cat([x @ w + b for w, b in zip(ws, bs)], dim=-1)
    assert torch.allclose(y, ref, atol=1e-5)
    o = ops.flash_attention_qkv(y, num_heads=4, causal=True)
    assert o.shape == (2, 8, 16)
