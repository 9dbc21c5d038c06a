"""Graph IR (reference include/nn/graph*.hpp).

The reference's graph machinery exists but in practice every shipped model
compiles to a single-edge graph wrapping one Sequential (reference
include/nn/example_models.hpp:49-71; SURVEY §2.6 "Scale caveat"). We keep
the same proportions: a small named-dataflow IR whose executor walks edges
in topological order, with branching handled inside blocks.

``GraphContext``'s param/grad slab packing (reference
include/nn/graph_context.hpp:37-88) maps to the PyTorch-ROCm caching
allocator + a flat ``zero_grads`` over parameters here.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import torch

from .layer import Layer, layer_from_config


class OpNode:
    """Wraps a Layer (reference include/nn/op_node.hpp:12)."""

    def __init__(self, layer: Layer, uid: int):
        self.layer = layer
        self.uid = uid


class Edge:
    """producers -> op -> consumers (reference include/nn/edge.hpp:7)."""

    def __init__(self, inputs: List[str], output: str, node: OpNode):
        self.inputs = list(inputs)
        self.output = output
        self.node = node


class Graph(torch.nn.Module):
    def __init__(self, name: str = "graph"):
        super().__init__()
        self.graph_name = name
        self.nodes = torch.nn.ModuleList()
        self._opnodes: List[OpNode] = []
        self.edges: List[Edge] = []

    # -- construction --------------------------------------------------------
    def add_node(self, layer: Layer) -> OpNode:
        node = OpNode(layer, uid=len(self._opnodes))
        self._opnodes.append(node)
        self.nodes.append(layer)
        return node

    def add_edge(self, inputs: List[str], output: str, node: OpNode):
        self.edges.append(Edge(inputs, output, node))

    # -- topo sort (Kahn; reference graph_builder.hpp:28-111) ----------------
    def compile(self):
        produced = {e.output: e for e in self.edges}
        order, resolved, pending = [], set(), list(self.edges)
        ext_inputs = {name for e in self.edges for name in e.inputs
                      if name not in produced}
        resolved |= ext_inputs
        while pending:
            progressed = False
            for e in list(pending):
                if all(i in resolved for i in e.inputs):
                    order.append(e)
                    resolved.add(e.output)
                    pending.remove(e)
                    progressed = True
            if not progressed:
                raise ValueError("graph has a cycle or unbound input")
        self.edges = order
        return self

    # -- execution -----------------------------------------------------------
    def forward(self, inputs: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        vals = dict(inputs)
        for e in self.edges:
            args = [vals[i] for i in e.inputs]
            vals[e.output] = e.node.layer(*args)
        return vals

    # -- config round-trip ---------------------------------------------------
    def get_config(self) -> Dict[str, Any]:
        return {
            "name": self.graph_name,
            "nodes": [n.layer.get_config() for n in self._opnodes],
            "edges": [{"inputs": e.inputs, "output": e.output,
                       "node": e.node.uid} for e in self.edges],
        }

    @classmethod
    def from_config(cls, cfg: Dict[str, Any]) -> "Graph":
        g = cls(cfg.get("name", "graph"))
        for node_cfg in cfg["nodes"]:
            g.add_node(layer_from_config(node_cfg))
        for e in cfg["edges"]:
            g.add_edge(e["inputs"], e["output"], g._opnodes[e["node"]])
        return g.compile()


class GraphBuilder:
    """Accumulates nodes/edges then compiles (reference graph_builder.hpp:28)."""

    def __init__(self, name: str = "graph"):
        self.graph = Graph(name)

    def add_layer(self, layer: Layer, inputs: List[str], output: str) -> "GraphBuilder":
        node = self.graph.add_node(layer)
        self.graph.add_edge(inputs, output, node)
        return self

    def build(self) -> Graph:
        return self.graph.compile()


class GraphExecutor:
    """Binds named inputs/outputs and runs forward/backward
    (reference include/nn/graph_executor.hpp:30-163)."""

    def __init__(self, graph: Graph):
        self.graph = graph

    def forward(self, inputs: Dict[str, torch.Tensor],
                outputs: List[str]) -> List[torch.Tensor]:
        vals = self.graph(inputs)
        return [vals[o] for o in outputs]

    def backward(self, outputs: List[torch.Tensor],
                 grads: List[torch.Tensor]):
        torch.autograd.backward(outputs, grads)

    def zero_grads(self):
        for p in self.graph.parameters():
            p.grad = None


def wrap_sequential(model: Layer, name: Optional[str] = None) -> Graph:
    """The reference's standard shape: one OpNode wrapping the whole model,
    wired "input"→"output" (reference example_models.hpp:49-71)."""
    g = Graph(name or getattr(model, "name", "model"))
    node = g.add_node(model)
    g.add_edge(["input"], "output", node)
    return g.compile()
