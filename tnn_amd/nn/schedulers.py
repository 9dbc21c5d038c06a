"""LR schedulers (reference include/nn/schedulers.hpp:27-619, all 12).

``step()`` is called once per optimizer update (or per epoch, matching how
the training loop wires it); ``step_metric(value)`` drives
ReduceLROnPlateau.
"""

from __future__ import annotations

import math
from typing import Any, Dict, List

from .optim import Optimizer


class Scheduler:
    _type = "noop"

    def __init__(self, optimizer: Optimizer):
        self.optimizer = optimizer
        self.base_lr = optimizer.lr
        self.t = 0

    def step(self):
        self.t += 1
        self.optimizer.lr = self.lr_at(self.t)

    def step_metric(self, metric: float):
        self.step()

    def lr_at(self, t: int) -> float:
        return self.base_lr

    def get_config(self) -> Dict[str, Any]:
        return {"type": self._type, **self.extra_config()}

    def extra_config(self) -> Dict[str, Any]:
        return {}


class NoOpScheduler(Scheduler):
    _type = "noop"


class StepLR(Scheduler):
    _type = "step"

    def __init__(self, optimizer, step_size: int = 10, gamma: float = 0.1):
        super().__init__(optimizer)
        self.step_size, self.gamma = step_size, gamma

    def lr_at(self, t):
        return self.base_lr * self.gamma ** (t // self.step_size)

    def extra_config(self):
        return {"step_size": self.step_size, "gamma": self.gamma}


class MultiStepLR(Scheduler):
    _type = "multistep"

    def __init__(self, optimizer, milestones: List[int] = (), gamma: float = 0.1):
        super().__init__(optimizer)
        self.milestones, self.gamma = sorted(milestones), gamma

    def lr_at(self, t):
        k = sum(1 for m in self.milestones if t >= m)
        return self.base_lr * self.gamma ** k

    def extra_config(self):
        return {"milestones": list(self.milestones), "gamma": self.gamma}


class ExponentialLR(Scheduler):
    _type = "exponential"

    def __init__(self, optimizer, gamma: float = 0.95):
        super().__init__(optimizer)
        self.gamma = gamma

    def lr_at(self, t):
        return self.base_lr * self.gamma ** t

    def extra_config(self):
        return {"gamma": self.gamma}


class CosineAnnealingLR(Scheduler):
    _type = "cosine"

    def __init__(self, optimizer, t_max: int = 100, eta_min: float = 0.0):
        super().__init__(optimizer)
        self.t_max, self.eta_min = t_max, eta_min

    def lr_at(self, t):
        return self.eta_min + (self.base_lr - self.eta_min) * \
            (1 + math.cos(math.pi * min(t, self.t_max) / self.t_max)) / 2

    def extra_config(self):
        return {"t_max": self.t_max, "eta_min": self.eta_min}


class CosineAnnealingWarmRestarts(Scheduler):
    _type = "cosine_warm_restarts"

    def __init__(self, optimizer, t0: int = 10, t_mult: int = 2, eta_min: float = 0.0):
        super().__init__(optimizer)
        self.t0, self.t_mult, self.eta_min = t0, t_mult, eta_min

    def lr_at(self, t):
        ti, tcur = self.t0, t
        while tcur >= ti:
            tcur -= ti
            ti *= self.t_mult
        return self.eta_min + (self.base_lr - self.eta_min) * \
            (1 + math.cos(math.pi * tcur / ti)) / 2

    def extra_config(self):
        return {"t0": self.t0, "t_mult": self.t_mult, "eta_min": self.eta_min}


class LinearWarmup(Scheduler):
    _type = "linear_warmup"

    def __init__(self, optimizer, warmup_steps: int = 100):
        super().__init__(optimizer)
        self.warmup_steps = warmup_steps

    def lr_at(self, t):
        return self.base_lr * min(1.0, t / max(1, self.warmup_steps))

    def extra_config(self):
        return {"warmup_steps": self.warmup_steps}


class WarmupCosineAnnealing(Scheduler):
    _type = "warmup_cosine"

    def __init__(self, optimizer, warmup_steps: int = 100, t_max: int = 1000,
                 eta_min: float = 0.0):
        super().__init__(optimizer)
        self.warmup_steps, self.t_max, self.eta_min = warmup_steps, t_max, eta_min

    def lr_at(self, t):
        if t < self.warmup_steps:
            return self.base_lr * t / max(1, self.warmup_steps)
        tt = min(t - self.warmup_steps, self.t_max - self.warmup_steps)
        span = max(1, self.t_max - self.warmup_steps)
        return self.eta_min + (self.base_lr - self.eta_min) * \
            (1 + math.cos(math.pi * tt / span)) / 2

    def extra_config(self):
        return {"warmup_steps": self.warmup_steps, "t_max": self.t_max,
                "eta_min": self.eta_min}


class ReduceLROnPlateau(Scheduler):
    _type = "reduce_on_plateau"

    def __init__(self, optimizer, factor: float = 0.1, patience: int = 10,
                 mode: str = "min", threshold: float = 1e-4):
        super().__init__(optimizer)
        self.factor, self.patience, self.mode, self.threshold = factor, patience, mode, threshold
        self.best = math.inf if mode == "min" else -math.inf
        self.bad = 0

    def step(self):
        self.t += 1  # plain step does nothing without a metric

    def step_metric(self, metric: float):
        self.t += 1
        improved = (metric < self.best - self.threshold) if self.mode == "min" \
            else (metric > self.best + self.threshold)
        if improved:
            self.best, self.bad = metric, 0
        else:
            self.bad += 1
            if self.bad > self.patience:
                self.optimizer.lr *= self.factor
                self.bad = 0

    def extra_config(self):
        return {"factor": self.factor, "patience": self.patience,
                "mode": self.mode, "threshold": self.threshold}


class PolynomialLR(Scheduler):
    _type = "polynomial"

    def __init__(self, optimizer, total_steps: int = 100, power: float = 1.0,
                 eta_min: float = 0.0):
        super().__init__(optimizer)
        self.total_steps, self.power, self.eta_min = total_steps, power, eta_min

    def lr_at(self, t):
        frac = 1 - min(t, self.total_steps) / self.total_steps
        return self.eta_min + (self.base_lr - self.eta_min) * frac ** self.power

    def extra_config(self):
        return {"total_steps": self.total_steps, "power": self.power,
                "eta_min": self.eta_min}


class OneCycleLR(Scheduler):
    _type = "one_cycle"

    def __init__(self, optimizer, max_lr: float = 0.1, total_steps: int = 100,
                 pct_start: float = 0.3, div_factor: float = 25.0,
                 final_div_factor: float = 1e4):
        super().__init__(optimizer)
        self.max_lr, self.total_steps = max_lr, total_steps
        self.pct_start, self.div_factor, self.final_div_factor = \
            pct_start, div_factor, final_div_factor

    def lr_at(self, t):
        up = int(self.total_steps * self.pct_start)
        lo = self.max_lr / self.div_factor
        fin = self.max_lr / self.final_div_factor
        if t <= up and up > 0:
            return lo + (self.max_lr - lo) * (1 - math.cos(math.pi * t / up)) / 2
        span = max(1, self.total_steps - up)
        tt = min(t - up, span)
        return fin + (self.max_lr - fin) * (1 + math.cos(math.pi * tt / span)) / 2

    def extra_config(self):
        return {"max_lr": self.max_lr, "total_steps": self.total_steps,
                "pct_start": self.pct_start, "div_factor": self.div_factor,
                "final_div_factor": self.final_div_factor}


SCHEDULERS = {c._type: c for c in
              [NoOpScheduler, StepLR, MultiStepLR, ExponentialLR, CosineAnnealingLR,
               CosineAnnealingWarmRestarts, LinearWarmup, WarmupCosineAnnealing,
               ReduceLROnPlateau, PolynomialLR, OneCycleLR]}


def scheduler_from_config(cfg: Dict[str, Any], optimizer: Optimizer) -> Scheduler:
    cfg = dict(cfg)
    cls = SCHEDULERS[cfg.pop("type")]
    return cls(optimizer, **cfg)
