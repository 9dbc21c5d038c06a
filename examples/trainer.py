#!/usr/bin/env python3
"""Single-process trainer (reference examples/trainer.cpp:16-81).

    python examples/trainer.py --model cifar100_wrn16_8 --dataset synthetic \
        --epochs 2 --batch-size 256 [--config configs/default_config.json]

Datasets: synthetic (no files needed), mnist/cifar10/cifar100 (need data
dirs in the reference's formats).
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnn_amd import models
from tnn_amd.data import DataLoaderFactory
from tnn_amd.nn import (TrainingConfig, train_model, CrossEntropyLoss, AdamW,
                        schedulers)
from tnn_amd.nn.layer import cast_compute_dtype, dtype_from_name
from tnn_amd.utils import EnvLoader


def main():
    EnvLoader.load()
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="cifar100_wrn16_8")
    p.add_argument("--dataset", default="synthetic")
    p.add_argument("--data-path", default="data")
    p.add_argument("--config", default=None)
    p.add_argument("--epochs", type=int, default=None)
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--lr", type=float, default=None)
    p.add_argument("--dtype", default="float32")
    args = p.parse_args()

    overrides = {k: v for k, v in [("epochs", args.epochs),
                                   ("batch_size", args.batch_size),
                                   ("learning_rate", args.lr)] if v is not None}
    cfg = (TrainingConfig.from_json(args.config, **overrides) if args.config
           else TrainingConfig.from_env(**overrides))

    model = models.create_model(args.model)
    dt = dtype_from_name(args.dtype)
    if dt != torch.float32:
        cast_compute_dtype(model, dt)

    shapes = {"mnist_cnn": ((28, 28, 1), 10), "cifar10_resnet9": ((32, 32, 3), 10),
              "cifar10_vgg": ((32, 32, 3), 10)}
    in_shape, classes = shapes.get(args.model, ((32, 32, 3), 100))
    if args.dataset == "synthetic":
        train_loader = DataLoaderFactory.create(
            "synthetic_image", shape=in_shape, num_classes=classes,
            num_samples=50 * cfg.batch_size, batch_size=cfg.batch_size)
        val_loader = DataLoaderFactory.create(
            "synthetic_image", shape=in_shape, num_classes=classes,
            num_samples=10 * cfg.batch_size, batch_size=cfg.batch_size, seed=1)
    else:
        train_loader = DataLoaderFactory.create(
            args.dataset, path=args.data_path, train=True,
            batch_size=cfg.batch_size)
        val_loader = DataLoaderFactory.create(
            args.dataset, path=args.data_path, train=False,
            batch_size=cfg.batch_size)

    opt = AdamW(model.parameters(), lr=cfg.learning_rate)
    sched = schedulers.CosineAnnealingLR(opt, t_max=cfg.epochs)
    result = train_model(model, train_loader, val_loader, CrossEntropyLoss(),
                         opt, sched, cfg)
    print("best val accuracy:", result["best_val_accuracy"])


if __name__ == "__main__":
    main()
