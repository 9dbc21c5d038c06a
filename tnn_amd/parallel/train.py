"""Distributed training loop (reference include/distributed/train.hpp:19-128:
train_semi_async_epoch / validate_semi_async_epoch / train_model wrapper).

The per-batch mechanics (micro-batch split, schedule, update barrier) live
in :class:`PipelineEngine`; this module is the epoch-level driver with the
same role as the reference's coordinator-side loop, plus per-stage
checkpointing (reference Worker SAVE_TO_FILE, worker.hpp:287).
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional

from ..utils.logging import get_logger
from .pipeline import PipelineEngine

log = get_logger("parallel.train")


def train_pipeline_epoch(engine: PipelineEngine, loader, epoch: int = 0,
                         log_interval: int = 10) -> Dict[str, float]:
    """One epoch over ``loader`` (every rank iterates the same loader —
    same seed — rank 0 consumes inputs, the last rank labels)."""
    total_loss, total_acc, batches = 0.0, 0.0, 0
    t0 = time.perf_counter()
    for i, (x, y) in enumerate(loader):
        stats = engine.train_batch(x, y)
        if engine.is_last:
            total_loss += stats["loss"]
            total_acc += stats["accuracy"]
        batches += 1
        if engine.is_last and log_interval and (i + 1) % log_interval == 0:
            log.info("epoch %d batch %d loss %.4f acc %.3f", epoch, i + 1,
                     total_loss / batches, total_acc / batches)
    dt = time.perf_counter() - t0
    stats = {"loss": total_loss / max(1, batches),
             "accuracy": total_acc / max(1, batches), "seconds": dt}
    return engine.broadcast_stats(stats)


def validate_pipeline_epoch(engine: PipelineEngine, loader) -> Dict[str, float]:
    total_loss, total_acc, batches = 0.0, 0.0, 0
    for x, y in loader:
        stats = engine.eval_batch(x, y)
        total_loss += stats["loss"]
        total_acc += stats["accuracy"]
        batches += 1
    return engine.broadcast_stats({"loss": total_loss / max(1, batches),
                                   "accuracy": total_acc / max(1, batches)})


def train_pipeline_model(engine: PipelineEngine, train_loader,
                         val_loader=None, epochs: int = 1,
                         snapshot_dir: Optional[str] = None,
                         log_interval: int = 10) -> Dict[str, Any]:
    """Epoch wrapper (reference train_model(Coordinator&, ...) :111)."""
    history = []
    best = -1.0
    for epoch in range(epochs):
        stats = train_pipeline_epoch(engine, train_loader, epoch, log_interval)
        entry = {"epoch": epoch, **{f"train_{k}": v for k, v in stats.items()}}
        if val_loader is not None:
            vstats = validate_pipeline_epoch(engine, val_loader)
            entry.update({f"val_{k}": v for k, v in vstats.items()})
            if snapshot_dir and vstats["accuracy"] > best:
                best = vstats["accuracy"]
                save_stage_checkpoint(engine, snapshot_dir)
        if engine.rank == 0:
            log.info("epoch %d: %s", epoch, {k: round(v, 4)
                     if isinstance(v, float) else v for k, v in entry.items()})
        history.append(entry)
    return {"history": history, "best_val_accuracy": best}


def save_stage_checkpoint(engine: PipelineEngine, directory: str):
    """Each rank writes its stage (reference Worker SAVE_TO_FILE: per-stage
    model files) in the standard archive format, optimizer state included."""
    os.makedirs(directory, exist_ok=True)
    from ..utils.checkpoint import save_checkpoint
    path = os.path.join(directory, f"stage_{engine.rank}.ckpt")
    save_checkpoint(engine.stage, engine.optimizer, path,
                    extra={"rank": engine.rank, "world": engine.world})
    engine.comm.barrier()


def load_stage_checkpoint(engine: PipelineEngine, directory: str):
    from ..utils.checkpoint import load_checkpoint
    path = os.path.join(directory, f"stage_{engine.rank}.ckpt")
    header = load_checkpoint(path, engine.stage, engine.optimizer)
    engine.comm.barrier()
    return header
