"""Checkpoint archive format round trips (reference archiver_test.cpp +
Graph::save_state/load_state)."""


import torch

from tnn_amd import models
from tnn_amd.nn import optim
from tnn_amd.utils.checkpoint import (save_model, load_model, save_checkpoint,
                                      load_checkpoint)


def test_model_save_load_bitwise(tmp_path):
    m = models.create_model("mnist_cnn")
    path = str(tmp_path / "m.ckpt")
    save_model(m, path)
    m2 = load_model(path)  # rebuild from embedded config
    sd1, sd2 = m.state_dict(), m2.state_dict()
    assert set(sd1) == set(sd2)
    for k in sd1:
        assert torch.equal(sd1[k], sd2[k]), k


def test_bf16_tensor_roundtrip(tmp_path):
    m = models.create_model("mnist_cnn", dtype=torch.bfloat16)
    path = str(tmp_path / "m.ckpt")
    save_model(m, path)
    m2 = load_model(path)
    for (k, a), (_, b) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert a.dtype == b.dtype
        assert torch.equal(a, b), k


def test_checkpoint_with_optimizer_state(tmp_path):
    m = models.create_model("mnist_cnn")
    o = optim.AdamW(m.parameters(), lr=1e-3)
    x = torch.randn(4, 28, 28, 1)
    m(x).sum().backward()
    o.step()
    path = str(tmp_path / "full.ckpt")
    save_checkpoint(m, o, path)

    # fresh optimizer that has NEVER stepped (the normal resume flow):
    # load_checkpoint must materialize its empty lazy state slots itself
    m2 = models.create_model("mnist_cnn")
    o2 = optim.AdamW(m2.parameters(), lr=1e-3)
    header = load_checkpoint(path, m2, o2)
    assert header["format"] == "tnn_amd.checkpoint.v1"
    assert o2.step_count == o.step_count
    for (k, a), (_, b) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert torch.equal(a, b), k
    t1s, t2s = o.state_tensors(), o2.state_tensors()
    assert len(t1s) == len(t2s) and len(t1s) > 0
    for (n1, t1), (n2, t2) in zip(t1s, t2s):
        assert n1 == n2
        assert torch.equal(t1, t2), n1

    # post-resume step must match a never-interrupted run (moments intact)
    o.zero_grad()
    o2.zero_grad()
    m2(x).sum().backward()
    o2.step()
    m(x).sum().backward()
    o.step()
    for (k, a), (_, b) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert torch.equal(a, b), k


def test_checkpoint_model_only_load_skips_opt_records(tmp_path):
    m = models.create_model("mnist_cnn")
    o = optim.AdamW(m.parameters(), lr=1e-3)
    m(torch.randn(2, 28, 28, 1)).sum().backward()
    o.step()
    path = str(tmp_path / "full.ckpt")
    save_checkpoint(m, o, path)
    m2 = models.create_model("mnist_cnn")
    load_checkpoint(path, m2, optimizer=None)  # must not raise
    for (k, a), (_, b) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert torch.equal(a, b), k
