#!/usr/bin/env python3
"""Pipeline-parallel trainer, one rank per GPU over RCCL/xGMI
(replaces reference examples/tcp_coordinator.cpp + tcp_worker.cpp /
roce_* binaries — the coordinator/worker pair becomes torchrun ranks).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
        --master-addr 127.0.0.1 examples/pipeline_trainer.py \
        --model cifar100_wrn16_8 --epochs 2
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnn_amd import models
from tnn_amd.data import DataLoaderFactory
from tnn_amd.nn import CrossEntropyLoss
from tnn_amd.parallel import (init_distributed, PipelineEngine,
                              train_pipeline_model)
from tnn_amd.utils import get_logger

log = get_logger("pipeline_trainer")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="cifar100_wrn16_8")
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--microbatches", type=int,
                   default=int(os.environ.get("NUM_MICROBATCHES", 4)))
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--dtype", default="bf16" if torch.cuda.is_available()
                   else "fp32")
    p.add_argument("--profile", default=None,
                   help="write a merged chrome trace here (rank 0)")
    p.add_argument("--snapshot-dir", default=None)
    args = p.parse_args()

    comm = init_distributed()
    model = models.create_model(args.model) if comm.rank == 0 else None
    engine = PipelineEngine(
        model, comm, input_shape=(32, 32, 3),
        num_microbatches=args.microbatches,
        criterion=CrossEntropyLoss(),
        optimizer_config={"type": "adamw", "lr": args.lr},
        scheduler_config={"type": "cosine", "t_max": args.epochs * 50},
        io_dtype=torch.bfloat16 if args.dtype == "bf16" else torch.float32)

    # every rank draws the same synthetic data (same seed) — rank 0
    # consumes inputs, the last rank labels
    loader = DataLoaderFactory.create(
        "synthetic_image", shape=(32, 32, 3), num_classes=100,
        num_samples=50 * args.batch_size, batch_size=args.batch_size, seed=7)
    val_loader = DataLoaderFactory.create(
        "synthetic_image", shape=(32, 32, 3), num_classes=100,
        num_samples=10 * args.batch_size, batch_size=args.batch_size, seed=8)
    if args.profile:
        engine.start_profiling()
    train_pipeline_model(engine, loader, val_loader, epochs=args.epochs,
                         snapshot_dir=args.snapshot_dir)
    if args.profile:
        merged = engine.gather_profiles()
        if merged is not None:
            merged.export_chrome_trace(args.profile)
            log.info("chrome trace written to %s", args.profile)
    comm.barrier()
    comm.destroy()


if __name__ == "__main__":
    main()
