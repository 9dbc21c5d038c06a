"""Optimizer/scheduler behavior (reference optimizers.hpp / schedulers.hpp)."""


import pytest
import torch

from tnn_amd.nn import optim, schedulers


def _quadratic_problem(opt_cls, steps=200, **kw):
    torch.manual_seed(0)
    target = torch.tensor([1.5, -2.0, 0.5])
    p = torch.nn.Parameter(torch.zeros(3))
    o = opt_cls([p], **kw)
    for _ in range(steps):
        loss = ((p - target) ** 2).sum()
        o.zero_grad()
        loss.backward()
        o.step()
    return p.detach(), target


@pytest.mark.parametrize("cls,kw", [
    (optim.SGD, {"lr": 0.05}),
    (optim.SGD, {"lr": 0.05, "momentum": 0.9}),
    (optim.SGD, {"lr": 0.05, "momentum": 0.9, "nesterov": True}),
    (optim.Adam, {"lr": 0.1}),
    (optim.AdamW, {"lr": 0.1, "weight_decay": 0.001}),
])
def test_optimizers_converge(cls, kw):
    p, target = _quadratic_problem(cls, **kw)
    assert torch.allclose(p, target, atol=0.05), (p, target)


def test_adam_matches_torch():
    torch.manual_seed(1)
    g = torch.randn(10)
    p1 = torch.nn.Parameter(torch.ones(10))
    p2 = torch.nn.Parameter(torch.ones(10))
    ours = optim.Adam([p1], lr=0.01)
    ref = torch.optim.Adam([p2], lr=0.01)
    for _ in range(5):
        p1.grad = g.clone()
        p2.grad = g.clone()
        ours.step()
        ref.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_adamw_matches_torch():
    torch.manual_seed(1)
    p1 = torch.nn.Parameter(torch.ones(10))
    p2 = torch.nn.Parameter(torch.ones(10))
    ours = optim.AdamW([p1], lr=0.01, weight_decay=0.1)
    ref = torch.optim.AdamW([p2], lr=0.01, weight_decay=0.1,
                            betas=(0.9, 0.999), eps=1e-8)
    for i in range(5):
        g = torch.randn(10)
        p1.grad = g.clone()
        p2.grad = g.clone()
        ours.step()
        ref.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_bf16_master_weights():
    p = torch.nn.Parameter(torch.ones(8, dtype=torch.bfloat16))
    o = optim.AdamW([p], lr=1e-3)
    for _ in range(3):
        p.grad = torch.full((8,), 0.1, dtype=torch.bfloat16)
        o.step()
    assert o.state[0]["master"].dtype == torch.float32
    assert p.dtype == torch.bfloat16
    assert (p < 1.0).all()


def test_all_schedulers_registered():
    assert set(schedulers.SCHEDULERS) == {
        "noop", "step", "multistep", "exponential", "cosine",
        "cosine_warm_restarts", "linear_warmup", "warmup_cosine",
        "reduce_on_plateau", "polynomial", "one_cycle"}


def test_step_and_cosine_lr():
    p = torch.nn.Parameter(torch.zeros(1))
    o = optim.SGD([p], lr=1.0)
    s = schedulers.StepLR(o, step_size=2, gamma=0.1)
    lrs = []
    for _ in range(4):
        s.step()
        lrs.append(o.lr)
    assert lrs == pytest.approx([1.0, 0.1, 0.1, 0.01])

    o2 = optim.SGD([p], lr=1.0)
    c = schedulers.CosineAnnealingLR(o2, t_max=10)
    for _ in range(10):
        c.step()
    assert o2.lr == pytest.approx(0.0, abs=1e-9)


def test_warmup_cosine():
    p = torch.nn.Parameter(torch.zeros(1))
    o = optim.SGD([p], lr=2.0)
    s = schedulers.WarmupCosineAnnealing(o, warmup_steps=5, t_max=15)
    s.step()
    assert o.lr == pytest.approx(2.0 / 5)
    for _ in range(4):
        s.step()
    assert o.lr == pytest.approx(2.0)
    for _ in range(10):
        s.step()
    assert o.lr == pytest.approx(0.0, abs=1e-9)


def test_reduce_on_plateau():
    p = torch.nn.Parameter(torch.zeros(1))
    o = optim.SGD([p], lr=1.0)
    s = schedulers.ReduceLROnPlateau(o, factor=0.5, patience=1)
    s.step_metric(1.0)
    s.step_metric(1.0)   # bad 1
    s.step_metric(1.0)   # bad 2 -> reduce
    assert o.lr == pytest.approx(0.5)


def test_scheduler_config_roundtrip():
    p = torch.nn.Parameter(torch.zeros(1))
    o = optim.AdamW([p], lr=0.1)
    s = schedulers.OneCycleLR(o, max_lr=1.0, total_steps=50)
    cfg = s.get_config()
    s2 = schedulers.scheduler_from_config(cfg, o)
    assert s2.get_config() == cfg
    ocfg = o.get_config()
    o2 = optim.optimizer_from_config(ocfg, [p])
    assert o2.get_config() == ocfg
