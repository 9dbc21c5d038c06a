"""Classification accuracy (reference include/nn/accuracy.hpp:31)."""

from __future__ import annotations

import torch


@torch.no_grad()
def accuracy(pred: torch.Tensor, target: torch.Tensor) -> float:
    """Fraction of rows whose argmax matches the target class."""
    p = pred.reshape(-1, pred.shape[-1]).argmax(-1)
    if target.dim() == pred.dim():            # one-hot targets
        t = target.reshape(-1, target.shape[-1]).argmax(-1)
    else:
        t = target.reshape(-1).long()
    return (p == t).float().mean().item()
