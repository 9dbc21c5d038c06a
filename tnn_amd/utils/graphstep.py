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


class GraphedTrainStep:
    """Whole-training-step hipGraph capture: grad-zeroing, forward, loss,
    backward and the fused optimizer update replay as ONE graph launch
    (the round-1 blockers are solved device-side: dropout Philox streams
    combine a capture-baked salt with an in-graph device counter, and the
    Adam bias correction reads the same counter instead of a baked-in
    host step — csrc/optim.hip, elementwise.hip, batchnorm.hip).

    Constraints: fixed shapes, constant LR (schedulers would freeze), the
    optimizer's python ``step_count`` stays at its capture value (the
    device counter is the truth; checkpoint via ``sync_step_count()``).
    """

    def __init__(self, model, criterion, optimizer, example_x, example_y,
                 warmup: int = 3):
        from ..ops.functional import set_graph_seed_ctr
        assert example_x.is_cuda
        self.model, self.criterion, self.optimizer = model, criterion, optimizer
        self.static_x = example_x.clone()
        self.static_y = example_y.clone()
        dev = example_x.device
        self.step_ctr = torch.zeros(1, dtype=torch.int64, device=dev)
        model.train()
        optimizer._step_dev = self.step_ctr
        # pack all grads into per-dtype slabs (the reference GraphContext's
        # grad-slab idea): grad zeroing is then ONE fill per dtype instead
        # of a launch per parameter (torch._foreach_zero_ fell back to
        # per-tensor fills on ROCm — ~34 launches/step on WRN)
        groups = {}
        for prm in optimizer.params:
            groups.setdefault(prm.dtype, []).append(prm)
        self._grad_slabs = []
        for dt, ps in groups.items():
            slab = torch.zeros(sum(pp.numel() for pp in ps), dtype=dt,
                               device=dev)
            off = 0
            for pp in ps:
                pp.grad = slab[off:off + pp.numel()].view_as(pp)
                off += pp.numel()
            self._grad_slabs.append(slab)
        set_graph_seed_ctr(self.step_ctr)
        try:
            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                for _ in range(warmup):
                    self._step_body()
            torch.cuda.current_stream().wait_stream(stream)
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.static_loss = self._step_body()
        finally:
            set_graph_seed_ctr(None)

    def _step_body(self):
        self.step_ctr.add_(1)
        # grads are views of the packed slabs: one fill per dtype
        for slab in self._grad_slabs:
            slab.zero_()
        out = self.model(self.static_x)
        loss = self.criterion(out, self.static_y)
        loss.backward()
        self.optimizer.step()
        return loss

    def __call__(self, x=None, y=None) -> torch.Tensor:
        """Run one training step; returns the (device) loss tensor."""
        if x is not None:
            self.static_x.copy_(x)
        if y is not None:
            self.static_y.copy_(y)
        self.graph.replay()
        return self.static_loss

    def sync_step_count(self):
        """Pull the device step counter back into the optimizer (one host
        sync; call before checkpointing)."""
        self.optimizer.step_count = int(self.step_ctr.item())
