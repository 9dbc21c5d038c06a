"""End-to-end single-process training (reference src/nn/train.cpp loop):
a small model must learn a synthetic problem."""

import torch

from tnn_amd import nn as tnn
from tnn_amd.data import SyntheticImageLoader
from tnn_amd.nn import TrainingConfig, train_model, CrossEntropyLoss, AdamW


def _tiny_cnn():
    return (tnn.LayerBuilder((8, 8, 1))
            .conv2d(8, 3, 3, 1, 1, 1, 1, True, "c1")
            .batchnorm(relu=True, name="b1")
            .maxpool2d(2, 2)
            .flatten()
            .dense(4, True, "fc")
            .build("tiny"))


def test_training_reduces_loss():
    torch.manual_seed(0)
    model = _tiny_cnn()
    # learnable synthetic task: class = quadrant with most energy
    n = 256
    x = torch.randn(n, 8, 8, 1).abs()
    y = torch.stack([x[:, :4, :4].sum((1, 2, 3)), x[:, :4, 4:].sum((1, 2, 3)),
                     x[:, 4:, :4].sum((1, 2, 3)), x[:, 4:, 4:].sum((1, 2, 3))],
                    dim=1).argmax(1)
    loader = [(x[i:i + 32], y[i:i + 32]) for i in range(0, n, 32)]
    crit = CrossEntropyLoss()
    opt = AdamW(model.parameters(), lr=5e-3)
    first_loss = None
    for epoch in range(15):
        total = 0.0
        for xb, yb in loader:
            out = model(xb)
            loss = crit(out, yb)
            opt.zero_grad()
            loss.backward()
            opt.step()
            total += loss.item()
        if first_loss is None:
            first_loss = total
    assert total < 0.6 * first_loss, (first_loss, total)


def test_train_model_api():
    torch.manual_seed(0)
    model = _tiny_cnn()
    loader = SyntheticImageLoader(shape=(8, 8, 1), num_classes=4,
                                  num_samples=64, batch_size=16)
    cfg = TrainingConfig(epochs=1, batch_size=16, learning_rate=1e-3,
                         device="cpu", log_interval=0)
    res = train_model(model, loader, val_loader=loader, cfg=cfg)
    assert len(res["history"]) == 1
    assert "val_accuracy" in res["history"][0]


def test_training_config_env(monkeypatch):
    monkeypatch.setenv("NUM_EPOCHS", "3")
    monkeypatch.setenv("NUM_MICROBATCHES", "8")
    cfg = TrainingConfig.from_env()
    assert cfg.epochs == 3
    assert cfg.num_microbatches == 8


def test_trainer_example_entrypoint(tmp_path):
    """examples/trainer.py end to end on CPU with a tiny synthetic run
    (the reference examples/trainer.cpp analog must stay runnable)."""
    import subprocess, sys, os
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cfg = tmp_path / "cfg.json"
    cfg.write_text('{"epochs": 1, "batch_size": 16, "learning_rate": 1e-3}')
    r = subprocess.run(
        [sys.executable, os.path.join(root, "examples", "trainer.py"),
         "--model", "mnist_cnn", "--dataset", "synthetic",
         "--config", str(cfg)],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "epoch" in (r.stdout + r.stderr).lower()


def test_inferencer_example_entrypoint():
    import subprocess, sys, os
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(root, "examples", "inferencer.py"),
         "--model", "mnist_cnn", "--dataset", "synthetic",
         "--batch-size", "32"],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-2000:]
