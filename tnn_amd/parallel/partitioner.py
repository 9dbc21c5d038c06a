"""Pipeline partitioners (reference include/partitioner/partitioner.hpp:50,
naive_partitioner.hpp:19-131).

``partition_model`` slices a Sequential's top-level layers into per-stage
Sequentials. Stages round-trip through config exactly like the reference's
``split()`` (partitioner.hpp:26-48), so a stage can be shipped to a rank as
JSON and re-instantiated there.

The FLOPs-weighted partitioner implements what the reference declared but
left dormant (weighted_partitioner.hpp:25-100 commented out): balance
stages by per-layer forward FLOPs from shape inference.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

from ..nn.blocks import Sequential


class Partitioner:
    def partition_model(self, model: Sequential, num_stages: int,
                        input_shape: Tuple[int, ...]) -> List[Sequential]:
        raise NotImplementedError

    @staticmethod
    def boundary_shapes(stages: Sequence[Sequential],
                        input_shape: Tuple[int, ...]) -> List[Tuple[int, ...]]:
        """Batchless activation shape at each stage boundary (len = stages+1)."""
        shapes = [tuple(input_shape)]
        for s in stages:
            shapes.append(s.output_shape(shapes[-1]))
        return shapes


class NaivePipelinePartitioner(Partitioner):
    """Equal (proportional) layer counts per stage
    (reference naive_partitioner.hpp:19-73)."""

    def __init__(self, proportions: Optional[Sequence[float]] = None):
        self.proportions = proportions

    def partition_model(self, model, num_stages, input_shape=None):
        n = len(model)
        props = self.proportions or [1.0 / num_stages] * num_stages
        assert len(props) == num_stages
        total = sum(props)
        counts = [max(1, round(p / total * n)) for p in props]
        # fix rounding so counts sum to n
        while sum(counts) > n:
            counts[counts.index(max(counts))] -= 1
        while sum(counts) < n:
            counts[counts.index(min(counts))] += 1
        stages, i = [], 0
        for s, c in enumerate(counts):
            stages.append(model.slice(i, i + c, name=f"{model.name}_stage{s}"))
            i += c
        return stages


class WeightedPipelinePartitioner(Partitioner):
    """FLOPs-balanced contiguous split via shape inference."""

    def partition_model(self, model, num_stages, input_shape):
        shape = tuple(input_shape)
        costs = []
        for layer in model:
            costs.append(max(1, layer.flops_per_item(shape)))
            shape = layer.output_shape(shape)
        total = sum(costs)
        target = total / num_stages
        stages, start, acc = [], 0, 0.0
        for i, c in enumerate(costs):
            acc += c
            remaining_layers = len(costs) - i - 1
            remaining_stages = num_stages - len(stages) - 1
            if (acc >= target and remaining_stages > 0 and
                    remaining_layers >= remaining_stages):
                stages.append(model.slice(start, i + 1,
                                          name=f"{model.name}_stage{len(stages)}"))
                start, acc = i + 1, 0.0
        stages.append(model.slice(start, len(costs),
                                  name=f"{model.name}_stage{len(stages)}"))
        while len(stages) < num_stages:  # degenerate tiny models
            stages.append(Sequential([], name=f"{model.name}_stage{len(stages)}"))
        return stages


class NaiveDataPartitioner:
    """Split a batch across data-parallel workers
    (reference naive_partitioner.hpp:79-131)."""

    def partition_input(self, x, num_parts: int):
        return list(x.chunk(num_parts, dim=0))


def partition_model(model: Sequential, num_stages: int,
                    input_shape: Tuple[int, ...],
                    strategy: str = "weighted") -> List[Sequential]:
    p: Partitioner = (WeightedPipelinePartitioner() if strategy == "weighted"
                      else NaivePipelinePartitioner())
    return p.partition_model(model, num_stages, input_shape)
