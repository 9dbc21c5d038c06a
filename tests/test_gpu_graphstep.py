"""Whole-training-step hipGraph capture: parity with eager steps and
fresh dropout randomness per replay (utils/graphstep.GraphedTrainStep)."""

import copy

import pytest
import torch

pytestmark = pytest.mark.gpu

from tnn_amd import models
from tnn_amd.nn import CrossEntropyLoss, AdamW
from tnn_amd.nn.layer import cast_compute_dtype
from tnn_amd.utils.graphstep import GraphedTrainStep

DEV = "cuda"


def _setup(name, dtype=torch.bfloat16, batch=64):
    torch.manual_seed(0)
    m = models.create_model(name)
    if dtype is not torch.float32:
        cast_compute_dtype(m, dtype)
    m = m.to(DEV)
    x = torch.randn(batch, 32, 32, 3, device=DEV)
    y = torch.randint(0, 10, (batch,), device=DEV)
    return m, x, y


def test_graphed_step_matches_eager():
    """Graphed warmup+replays track the eager trajectory (fp32,
    dropout-free model). Exact parity is unattainable — the fused BN
    statistics accumulate with atomics, so even eager-vs-eager runs
    differ at the ulp level and diverge chaotically over many steps on a
    memorized batch — so this checks a short horizon with loose bounds."""
    m1, x, y = _setup("cifar10_resnet9", dtype=torch.float32)
    m2 = copy.deepcopy(m1)
    crit = CrossEntropyLoss()
    o1 = AdamW(m1.parameters(), lr=1e-3)
    o2 = AdamW(m2.parameters(), lr=1e-3)

    m2.train()
    for _ in range(1 + 2):
        o2.zero_grad()
        loss2 = crit(m2(x), y)
        loss2.backward()
        o2.step()

    g = GraphedTrainStep(m1, crit, o1, x, y, warmup=1)
    for _ in range(2):
        loss1 = g()
    torch.cuda.synchronize()
    assert int(g.step_ctr.item()) == 3
    assert abs(loss1.item() - loss2.item()) < 0.1, (loss1.item(), loss2.item())
    n_close = sum(torch.allclose(a.float(), b.float(), atol=2e-2)
                  for (k, a), (_, b) in zip(m1.state_dict().items(),
                                            m2.state_dict().items()))
    total = len(m1.state_dict())
    assert n_close >= total - 2, (n_close, total)


def test_graphed_step_loss_decreases_with_dropout():
    """WRN (has fused BN dropout): replays draw fresh Philox streams and
    the loss goes down over replays on a fixed batch."""
    m, x, y = _setup("cifar100_wrn16_8", batch=128)
    y = torch.randint(0, 100, (128,), device=DEV)
    crit = CrossEntropyLoss()
    opt = AdamW(m.parameters(), lr=3e-3)
    g = GraphedTrainStep(m, crit, opt, x, y, warmup=2)
    first = g().item()
    for _ in range(20):
        loss = g()
    torch.cuda.synchronize()
    assert loss.item() < first, (first, loss.item())


def test_graphed_dropout_masks_vary_per_replay():
    from tnn_amd.ops import functional as Fn
    from tnn_amd import _C
    ext = _C.ext()
    ctr = torch.zeros(1, dtype=torch.int64, device=DEV)
    x = torch.ones(4096, device=DEV, dtype=torch.bfloat16)
    _, mask1 = ext.dropout_fwd(x, 0.5, 1234, ctr)
    ctr.add_(1)
    _, mask2 = ext.dropout_fwd(x, 0.5, 1234, ctr)
    assert not torch.equal(mask1, mask2)
    ctr.fill_(0)
    _, mask3 = ext.dropout_fwd(x, 0.5, 1234, ctr)
    assert torch.equal(mask1, mask3)  # counter value determines the stream


def test_conv_stats_ineligible_shape_computes_y():
    """Regression: K > 2304 (512-ch conv) is ineligible for the fused-stats
    db kernel; conv2d_fwd_stats must still compute y (round-1 latent bug
    returned uninitialized memory here, wrecking WRN/resnet9 training)."""
    from tnn_amd import _C
    ext = _C.ext()
    torch.manual_seed(1)
    x = torch.randn(32, 8, 8, 512, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(3, 3, 512, 512, dtype=torch.bfloat16, device=DEV) * 0.05
    for _ in range(3):
        y, stats = ext.conv2d_fwd_stats(x, w, None, 1, 1, 1, 1)
        y2 = ext.conv2d_fwd(x, w, None, 1, 1, 1, 1, False)
        assert torch.equal(y, y2)


def test_bf16_training_dynamics_sane():
    """bf16 resnet9 with the fused peepholes memorizes a fixed batch —
    the end-to-end training-dynamics check round 1 lacked."""
    torch.manual_seed(0)
    m = models.create_model("cifar10_resnet9")
    cast_compute_dtype(m, torch.bfloat16)
    m = m.to(DEV).train()
    x = torch.randn(64, 32, 32, 3, device=DEV)
    y = torch.randint(0, 10, (64,), device=DEV)
    crit = CrossEntropyLoss()
    opt = AdamW(m.parameters(), lr=1e-3)
    losses = []
    for _ in range(25):
        opt.zero_grad()
        l = crit(m(x), y)
        l.backward()
        opt.step()
        losses.append(l.item())
    assert all(v == v for v in losses), "NaN loss"
    assert losses[-1] < 0.7 * losses[0], losses[::6]


@pytest.mark.gpu
def test_bn_passthrough_residual_grads():
    """BN residual passthrough (pre-activation block junction): grads
    must match the unfused two-consumer formulation."""
    import torch
    from tnn_amd import ops
    torch.manual_seed(11)
    x = torch.randn(64, 8, 8, 32, dtype=torch.bfloat16, device="cuda")
    g = torch.randn(32, device="cuda").abs() + 0.5
    b = torch.randn(32, device="cuda")

    def run(passthrough):
        rm = torch.zeros(32, device="cuda")
        rv = torch.ones(32, device="cuda")
        xa = x.clone().requires_grad_()
        if passthrough:
            y, xr = ops.batch_norm_act(xa, g, b, rm, rv, True, relu=True,
                                       passthrough=True)
        else:
            y = ops.batch_norm_act(xa, g, b, rm, rv, True, relu=True)
            xr = xa
        out = y * 0.7 + xr  # join: both consumers
        out.float().square().mean().backward()
        return out.detach(), xa.grad

    o1, g1 = run(True)
    o2, g2 = run(False)
    assert (o1.float() - o2.float()).abs().max().item() < 1e-2
    assert (g1.float() - g2.float()).abs().max().item() < 2e-3


@pytest.mark.gpu
def test_residual_block_passthrough_matches_eval_math():
    """ResidualBlock's BN-passthrough fast path (train) vs the plain
    composition on the same weights."""
    import torch
    from tnn_amd.nn.builder import LayerBuilder
    torch.manual_seed(12)
    m = (LayerBuilder(input_shape=(8, 8, 32), dtype=torch.float32)
         .wide_residual_block(32, 32, 1, 0.0, "wb").build("m"))
    from tnn_amd.nn.layer import cast_compute_dtype
    cast_compute_dtype(m, torch.bfloat16)
    m.to("cuda").train()
    x = torch.randn(16, 8, 8, 32, dtype=torch.bfloat16, device="cuda",
                    requires_grad=True)
    y = m(x)
    y.float().square().mean().backward()
    assert torch.isfinite(y.float()).all()
    assert x.grad is not None and torch.isfinite(x.grad.float()).all()
