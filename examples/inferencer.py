#!/usr/bin/env python3
"""Validation-only runner (reference examples/inferencer.cpp:13)."""

import argparse
import os
import sys


sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnn_amd import models
from tnn_amd.data import DataLoaderFactory
from tnn_amd.nn import TrainingConfig, validate_model, CrossEntropyLoss
from tnn_amd.utils.checkpoint import load_model


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="cifar100_wrn16_8")
    p.add_argument("--snapshot", default=None, help="checkpoint path")
    p.add_argument("--dataset", default="synthetic")
    p.add_argument("--data-path", default="data")
    p.add_argument("--batch-size", type=int, default=256)
    args = p.parse_args()

    if args.snapshot:
        model = load_model(args.snapshot)
    else:
        model = models.create_model(args.model)
    if args.dataset == "synthetic":
        shapes = {"mnist_cnn": ((28, 28, 1), 10),
                  "cifar10_resnet9": ((32, 32, 3), 10),
                  "cifar10_vgg": ((32, 32, 3), 10)}
        in_shape, classes = shapes.get(args.model, ((32, 32, 3), 100))
        loader = DataLoaderFactory.create("synthetic_image", shape=in_shape,
                                          num_classes=classes,
                                          num_samples=2048,
                                          batch_size=args.batch_size)
    else:
        loader = DataLoaderFactory.create(args.dataset, path=args.data_path,
                                          train=False,
                                          batch_size=args.batch_size)
    cfg = TrainingConfig(batch_size=args.batch_size)
    stats = validate_model(model, loader, CrossEntropyLoss(), cfg)
    print(f"val loss {stats['loss']:.4f} acc {stats['accuracy']:.4f}")


if __name__ == "__main__":
    main()
