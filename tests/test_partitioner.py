"""Partitioner math (reference unit_tests/partitioner_test.cpp)."""

import torch

from tnn_amd import models
from tnn_amd.parallel.partitioner import (NaivePipelinePartitioner,
                                          WeightedPipelinePartitioner,
                                          NaiveDataPartitioner, Partitioner)


def _model():
    return models.create_model("cifar100_wrn16_8")


def test_naive_partition_counts():
    m = _model()
    stages = NaivePipelinePartitioner().partition_model(m, 4, (32, 32, 3))
    assert len(stages) == 4
    assert sum(len(s) for s in stages) == len(m)


def test_naive_proportions():
    m = _model()
    stages = NaivePipelinePartitioner([0.5, 0.5]).partition_model(m, 2, None)
    assert len(stages) == 2
    assert abs(len(stages[0]) - len(stages[1])) <= 1


def test_weighted_partition_balances_flops():
    m = _model()
    stages = WeightedPipelinePartitioner().partition_model(m, 2, (32, 32, 3))
    shapes = Partitioner.boundary_shapes(stages, (32, 32, 3))
    assert shapes[0] == (32, 32, 3)
    assert shapes[-1] == (100,)
    f0 = stages[0].flops_per_item((32, 32, 3))
    f1 = stages[1].flops_per_item(shapes[1])
    total = f0 + f1
    assert 0.25 < f0 / total < 0.75, (f0, f1)  # naive split is ~0.9/0.1


def test_stage_composition_equals_model():
    m = _model()
    m.eval()
    stages = WeightedPipelinePartitioner().partition_model(m, 3, (32, 32, 3))
    x = torch.randn(2, 32, 32, 3)
    y = x
    for s in stages:
        s.eval()
        y = s(y)
    assert torch.allclose(y, m(x), atol=1e-6)


def test_stage_config_roundtrip():
    from tnn_amd.nn.layer import layer_from_config
    m = _model()
    stages = WeightedPipelinePartitioner().partition_model(m, 4, (32, 32, 3))
    for s in stages:
        cfg = s.get_config()
        rebuilt = layer_from_config(cfg)
        assert rebuilt.get_config() == cfg


def test_data_partitioner():
    x = torch.randn(10, 3)
    parts = NaiveDataPartitioner().partition_input(x, 2)
    assert len(parts) == 2 and parts[0].shape[0] == 5
