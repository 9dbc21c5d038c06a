"""Losses (reference include/nn/loss.hpp:24-464).

Each loss maps (prediction, target) → scalar mean loss; gradients come from
autograd (the GPU cross-entropy path runs the fused logsoftmax HIP kernels,
reference src/nn/loss_impl/cuda/loss_ops.cu:76,150).

Targets may be int64 class ids or one-hot/prob rows (the reference's
OpenWebText loader emits one-hot [B, S, vocab]).
"""

from __future__ import annotations

from typing import Any, Dict

import torch
import torch.nn.functional as F

from ..ops import pointwise_loss, softmax_cross_entropy


class Loss:
    _type = "loss"

    def __call__(self, pred: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def get_config(self) -> Dict[str, Any]:
        return {"type": self._type, **self.extra_config()}

    def extra_config(self) -> Dict[str, Any]:
        return {}


class CrossEntropyLoss(Loss):
    """Fused logsoftmax+CE from raw logits (reference loss.hpp:68 with
    from_logits=true + LossFactory::create_logsoftmax_crossentropy :464)."""

    _type = "cross_entropy"

    def __call__(self, pred, target):
        if target.dtype in (torch.int64, torch.int32):
            return softmax_cross_entropy(pred, target.long())
        # float targets (one-hot, label-smoothed or genuinely soft): exact
        # soft cross-entropy -(t * log_softmax(p)).sum(-1) — an argmax here
        # would silently discard non-one-hot probability mass
        p2 = pred.reshape(-1, pred.shape[-1]).float()
        t2 = target.reshape(-1, target.shape[-1]).float()
        return -(t2 * F.log_softmax(p2, dim=-1)).sum(-1).mean()


class MSELoss(Loss):
    """Fused GPU reduce kernel (reference loss_ops.cu:308-330)."""

    _type = "mse"

    def __call__(self, pred, target):
        return pointwise_loss(pred, target, "mse")


class MAELoss(Loss):
    _type = "mae"

    def __call__(self, pred, target):
        return pointwise_loss(pred, target, "mae")


class HuberLoss(Loss):
    _type = "huber"

    def __init__(self, delta: float = 1.0):
        self.delta = delta

    def __call__(self, pred, target):
        return pointwise_loss(pred, target, "huber", self.delta)

    def extra_config(self):
        return {"delta": self.delta}


_LOSSES = {c._type: c for c in [CrossEntropyLoss, MSELoss, MAELoss, HuberLoss]}


def loss_from_config(cfg: Dict[str, Any]) -> Loss:
    cfg = dict(cfg)
    cls = _LOSSES[cfg.pop("type")]
    return cls(**cfg)
