"""Single-process training/validation loops
(reference include/nn/train.hpp:80-86, src/nn/train.cpp:129-272).

The hot loop: get_batch → forward → loss → backward → (every
``grad_accum``) optimizer step + zero + scheduler — one stream, no
host sync except the per-batch loss read (and that only when logging).
"""

from __future__ import annotations

import dataclasses
import json
import os
import time
from typing import Any, Dict, Optional

import torch

from .accuracy import accuracy
from .losses import Loss, CrossEntropyLoss
from .optim import Optimizer
from .schedulers import Scheduler, NoOpScheduler
from ..utils.logging import get_logger

log = get_logger("train")


@dataclasses.dataclass
class TrainingConfig:
    """reference include/nn/train.hpp:45-73 + src/nn/train.cpp:50-127
    (env- or JSON-loadable)."""

    epochs: int = 10
    batch_size: int = 128
    learning_rate: float = 1e-3
    grad_accum_steps: int = 1
    num_microbatches: int = 4
    device: str = "cuda" if torch.cuda.is_available() else "cpu"
    io_dtype: str = "float32"
    log_interval: int = 50
    snapshot_dir: str = "model_snapshots"
    save_best: bool = False

    @classmethod
    def from_env(cls, **overrides) -> "TrainingConfig":
        def get(name, cast, default):
            v = os.environ.get(name)
            return cast(v) if v is not None else default
        cfg = cls(
            epochs=get("NUM_EPOCHS", int, cls.epochs),
            batch_size=get("BATCH_SIZE", int, cls.batch_size),
            learning_rate=get("LEARNING_RATE", float, cls.learning_rate),
            grad_accum_steps=get("GRAD_ACCUM_STEPS", int, cls.grad_accum_steps),
            num_microbatches=get("NUM_MICROBATCHES", int, cls.num_microbatches),
            device=get("DEVICE_TYPE", str, cls.device),
            io_dtype=get("IO_DTYPE", str, cls.io_dtype),
        )
        return dataclasses.replace(cfg, **overrides)

    @classmethod
    def from_json(cls, path: str, **overrides) -> "TrainingConfig":
        with open(path) as f:
            data = json.load(f)
        known = {f.name for f in dataclasses.fields(cls)}
        cfg = cls(**{k: v for k, v in data.items() if k in known})
        return dataclasses.replace(cfg, **overrides)


def train_epoch(model, loader, criterion: Loss, optimizer: Optimizer,
                scheduler: Optional[Scheduler], cfg: TrainingConfig,
                epoch: int = 0) -> Dict[str, float]:
    model.train()
    device = torch.device(cfg.device)
    total_loss, total_acc, batches = 0.0, 0.0, 0
    t0 = time.perf_counter()
    optimizer.zero_grad()
    for i, (x, y) in enumerate(loader):
        x, y = x.to(device), y.to(device)
        out = model(x)
        loss = criterion(out, y) / cfg.grad_accum_steps
        loss.backward()
        if (i + 1) % cfg.grad_accum_steps == 0:
            optimizer.step()
            optimizer.zero_grad()
        total_loss += loss.item() * cfg.grad_accum_steps
        total_acc += accuracy(out, y)
        batches += 1
        if cfg.log_interval and (i + 1) % cfg.log_interval == 0:
            log.info("epoch %d batch %d loss %.4f acc %.4f",
                     epoch, i + 1, total_loss / batches, total_acc / batches)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if scheduler is not None:
        scheduler.step()
    return {"loss": total_loss / max(1, batches),
            "accuracy": total_acc / max(1, batches),
            "seconds": dt}


@torch.no_grad()
def validate_model(model, loader, criterion: Loss, cfg: TrainingConfig) -> Dict[str, float]:
    model.eval()
    device = torch.device(cfg.device)
    total_loss, total_acc, batches = 0.0, 0.0, 0
    for x, y in loader:
        x, y = x.to(device), y.to(device)
        out = model(x)
        total_loss += criterion(out, y).item()
        total_acc += accuracy(out, y)
        batches += 1
    return {"loss": total_loss / max(1, batches),
            "accuracy": total_acc / max(1, batches)}


def train_model(model, train_loader, val_loader=None,
                criterion: Optional[Loss] = None,
                optimizer: Optional[Optimizer] = None,
                scheduler: Optional[Scheduler] = None,
                cfg: Optional[TrainingConfig] = None) -> Dict[str, Any]:
    """Epoch loop with best-val snapshotting (reference src/nn/train.cpp:219-272)."""
    cfg = cfg or TrainingConfig()
    criterion = criterion or CrossEntropyLoss()
    if optimizer is None:
        from .optim import Adam
        optimizer = Adam(model.parameters(), lr=cfg.learning_rate)
    scheduler = scheduler or NoOpScheduler(optimizer)
    model.to(torch.device(cfg.device))
    history = []
    best_val = -1.0
    for epoch in range(cfg.epochs):
        stats = train_epoch(model, train_loader, criterion, optimizer, scheduler,
                            cfg, epoch)
        entry = {"epoch": epoch, **{f"train_{k}": v for k, v in stats.items()}}
        if val_loader is not None:
            vstats = validate_model(model, val_loader, criterion, cfg)
            entry.update({f"val_{k}": v for k, v in vstats.items()})
            if vstats["accuracy"] > best_val:
                best_val = vstats["accuracy"]
                if cfg.save_best:
                    from ..utils.checkpoint import save_model
                    os.makedirs(cfg.snapshot_dir, exist_ok=True)
                    save_model(model, os.path.join(
                        cfg.snapshot_dir, getattr(model, "name", "model")))
        log.info("epoch %d done: %s", epoch,
                 {k: round(v, 4) if isinstance(v, float) else v
                  for k, v in entry.items()})
        history.append(entry)
    return {"history": history, "best_val_accuracy": best_val}
