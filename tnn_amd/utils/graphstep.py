"""hipGraph capture for fixed-shape inference (HIP graphs are the MI355X
answer to launch-bound serving loops; torch.cuda.CUDAGraph is hipGraph on
ROCm).

``GraphedInference`` captures one eval-mode forward over static buffers
and replays it per call — every kernel launch in the model collapses into
a single graph launch. Restricted to inference on purpose: the training
step contains host-side RNG seeding (dropout) and python-side Adam step
counts that a captured graph would freeze (documented round-2 work).
"""

from __future__ import annotations


import torch


class GraphedInference:
    def __init__(self, model: torch.nn.Module, example_input: torch.Tensor,
                 warmup: int = 3):
        assert example_input.is_cuda, "graph capture needs GPU tensors"
        self.model = model
        model.eval()
        self.static_in = example_input.clone()
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            with torch.no_grad():
                for _ in range(warmup):
                    out = model(self.static_in)
        torch.cuda.current_stream().wait_stream(stream)
        self.graph = torch.cuda.CUDAGraph()
        with torch.no_grad():
            with torch.cuda.graph(self.graph):
                self.static_out = model(self.static_in)

    @torch.no_grad()
    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        self.static_in.copy_(x)
        self.graph.replay()
        return self.static_out
