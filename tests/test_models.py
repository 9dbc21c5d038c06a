"""Model zoo builds + forward shapes (reference example_models registry)."""

import pytest
import torch

from tnn_amd import models


def test_registry_complete():
    names = models.model_names()
    for required in ["mnist_cnn", "cifar10_vgg", "cifar10_resnet9",
                     "cifar100_resnet18", "cifar100_wrn16_8",
                     "tiny_imagenet_resnet18", "tiny_imagenet_wrn16_8",
                     "tiny_imagenet_resnet50", "imagenet_resnet50",
                     "tiny_imagenet_vit", "tiny_imagenet_flash_vit",
                     "gpt2_small", "gpt2_medium", "gpt2_large",
                     "flash_gpt2_small", "flash_gpt2_medium", "flash_gpt2_large"]:
        assert required in names, required


@pytest.mark.parametrize("name,in_shape,out_dim", [
    ("mnist_cnn", (28, 28, 1), 10),
    ("cifar10_resnet9", (32, 32, 3), 10),
    ("cifar100_wrn16_8", (32, 32, 3), 100),
    ("cifar100_resnet18", (32, 32, 3), 100),
])
def test_model_forward(name, in_shape, out_dim):
    m = models.create_model(name)
    m.eval()
    x = torch.randn(2, *in_shape)
    y = m(x)
    assert y.shape == (2, out_dim)
    assert m.output_shape(in_shape) == (out_dim,)


def test_wrn16_8_structure():
    """The headline model: 2.77M params expected for WRN-16-8/CIFAR-100
    (standard WRN-16-8 is ~11M at width 8 on paper counting; check finite)."""
    m = models.create_model("cifar100_wrn16_8")
    n = m.param_count()
    assert 5e6 < n < 20e6, n  # WRN-16-8 ≈ 11M params
    y = m(torch.randn(2, 32, 32, 3))
    assert y.shape == (2, 100)
    assert torch.isfinite(y).all()


def test_vit_forward():
    m = models.create_model("tiny_imagenet_vit")
    m.eval()
    y = m(torch.randn(2, 64, 64, 3))
    assert y.shape == (2, 200)


def test_model_config_roundtrip():
    from tnn_amd.nn.layer import layer_from_config
    m = models.create_model("cifar10_resnet9")
    cfg = m.get_config()
    m2 = layer_from_config(cfg)
    assert m2.get_config() == cfg
    assert m2.param_count() == m.param_count()
