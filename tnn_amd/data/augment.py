"""Image augmentations on NHWC float batches
(reference include/data_augmentation/augmentation.hpp:17,48,107 — 9
augmentations + strategy/builder)."""

from __future__ import annotations

import math
from typing import List, Optional

import numpy as np
import torch
import torch.nn.functional as F


class Augmentation:
    prob: float = 1.0

    def apply(self, x: torch.Tensor, rng: np.random.Generator) -> torch.Tensor:
        raise NotImplementedError

    def __call__(self, x, rng):
        if self.prob >= 1.0 or rng.random() < self.prob:
            return self.apply(x, rng)
        return x


class HorizontalFlip(Augmentation):
    def __init__(self, prob: float = 0.5):
        self.prob = prob

    def apply(self, x, rng):
        sel = torch.from_numpy(rng.random(x.shape[0]) < 0.5)
        x = x.clone()
        x[sel] = torch.flip(x[sel], dims=[2])
        return x


class RandomCrop(Augmentation):
    def __init__(self, padding: int = 4, prob: float = 1.0):
        self.padding, self.prob = padding, prob

    def apply(self, x, rng):
        n, h, w, c = x.shape
        p = self.padding
        xp = F.pad(x.permute(0, 3, 1, 2), (p, p, p, p)).permute(0, 2, 3, 1)
        out = torch.empty_like(x)
        offs = rng.integers(0, 2 * p + 1, size=(n, 2))
        for i in range(n):
            oy, ox = offs[i]
            out[i] = xp[i, oy:oy + h, ox:ox + w]
        return out


class Rotate(Augmentation):
    def __init__(self, max_degrees: float = 15.0, prob: float = 0.5):
        self.max_degrees, self.prob = max_degrees, prob

    def apply(self, x, rng):
        n = x.shape[0]
        deg = torch.from_numpy(
            rng.uniform(-self.max_degrees, self.max_degrees, n)).float()
        rad = deg * math.pi / 180
        cos, sin = torch.cos(rad), torch.sin(rad)
        theta = torch.zeros(n, 2, 3)
        theta[:, 0, 0], theta[:, 0, 1] = cos, -sin
        theta[:, 1, 0], theta[:, 1, 1] = sin, cos
        xn = x.permute(0, 3, 1, 2)
        grid = F.affine_grid(theta, xn.shape, align_corners=False)
        return F.grid_sample(xn, grid, align_corners=False).permute(0, 2, 3, 1)


class Brightness(Augmentation):
    def __init__(self, max_delta: float = 0.2, prob: float = 0.5):
        self.max_delta, self.prob = max_delta, prob

    def apply(self, x, rng):
        delta = torch.from_numpy(
            rng.uniform(-self.max_delta, self.max_delta, x.shape[0])).float()
        return x + delta.view(-1, 1, 1, 1)


class Contrast(Augmentation):
    def __init__(self, max_factor: float = 0.2, prob: float = 0.5):
        self.max_factor, self.prob = max_factor, prob

    def apply(self, x, rng):
        f = torch.from_numpy(
            rng.uniform(1 - self.max_factor, 1 + self.max_factor, x.shape[0])).float()
        mean = x.mean(dim=(1, 2, 3), keepdim=True)
        return (x - mean) * f.view(-1, 1, 1, 1) + mean


class GaussianNoise(Augmentation):
    def __init__(self, std: float = 0.05, prob: float = 0.5):
        self.std, self.prob = std, prob

    def apply(self, x, rng):
        return x + self.std * torch.from_numpy(
            rng.standard_normal(tuple(x.shape)).astype(np.float32))


class Cutout(Augmentation):
    def __init__(self, size: int = 8, prob: float = 0.5):
        self.size, self.prob = size, prob

    def apply(self, x, rng):
        n, h, w, _ = x.shape
        x = x.clone()
        ys = rng.integers(0, max(1, h - self.size), n)
        xs = rng.integers(0, max(1, w - self.size), n)
        for i in range(n):
            x[i, ys[i]:ys[i] + self.size, xs[i]:xs[i] + self.size, :] = 0
        return x


class Normalize(Augmentation):
    def __init__(self, mean, std):
        self.mean = torch.tensor(mean).view(1, 1, 1, -1)
        self.std = torch.tensor(std).view(1, 1, 1, -1)

    def apply(self, x, rng):
        return (x - self.mean.to(x.dtype)) / self.std.to(x.dtype)


class AugmentationStrategy:
    """Ordered augmentation pipeline + builder
    (reference augmentation.hpp:48,107)."""

    def __init__(self, augmentations: Optional[List[Augmentation]] = None):
        self.augmentations = augmentations or []

    def add(self, aug: Augmentation) -> "AugmentationStrategy":
        self.augmentations.append(aug)
        return self

    def __call__(self, x: torch.Tensor, rng: np.random.Generator) -> torch.Tensor:
        for aug in self.augmentations:
            x = aug(x, rng)
        return x
