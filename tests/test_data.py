"""Data loaders / augmentation / tokenizer (reference §2.7)."""

import numpy as np
import torch

from tnn_amd.data import (CIFAR100Loader, OpenWebTextLoader, DataLoaderFactory,
                          AugmentationStrategy, HorizontalFlip, RandomCrop,
                          Cutout, Normalize, Tokenizer)


def test_synthetic_image_loader():
    l = DataLoaderFactory.create("synthetic_image", shape=(32, 32, 3),
                                 num_classes=100, num_samples=128, batch_size=32)
    batches = list(l)
    assert len(batches) == 4
    x, y = batches[0]
    assert x.shape == (32, 32, 32, 3)
    assert y.dtype == torch.int64
    assert y.max() < 100


def test_cifar100_binary_format(tmp_path):
    # write two fake CIFAR-100 records (coarse label, fine label, 3072 bytes)
    rec = np.zeros((2, 2 + 3072), dtype=np.uint8)
    rec[0, 1] = 7
    rec[1, 1] = 42
    rec[:, 2:] = np.arange(3072, dtype=np.uint8).reshape(1, -1) % 255
    (tmp_path / "train.bin").write_bytes(rec.tobytes())
    l = CIFAR100Loader(str(tmp_path), train=True, batch_size=2, shuffle=False)
    x, y = next(iter(l))
    assert x.shape == (2, 32, 32, 3)
    assert y.tolist() == [7, 42]
    # CHW uint8 -> NHWC float conversion: channel 0 pixel (0,0) was byte 0
    assert x[0, 0, 0, 0].item() == 0.0


def test_openwebtext_loader(tmp_path):
    tokens = np.arange(10000, dtype=np.uint16)
    p = tmp_path / "train.bin"
    tokens.tofile(p)
    l = OpenWebTextLoader(str(p), seq_len=16, samples_per_epoch=8, batch_size=4)
    x, y = next(iter(l))
    assert x.shape == (4, 16)
    assert torch.equal(y[:, :-1], x[:, 1:])  # next-token targets


def test_augmentations_preserve_shape():
    x = torch.randn(4, 32, 32, 3)
    rng = np.random.default_rng(0)
    strat = (AugmentationStrategy()
             .add(HorizontalFlip(1.0))
             .add(RandomCrop(4))
             .add(Cutout(8, prob=1.0))
             .add(Normalize([0.5, 0.5, 0.5], [0.25, 0.25, 0.25])))
    y = strat(x, rng)
    assert y.shape == x.shape


def test_tokenizer_roundtrip(tmp_path):
    t = Tokenizer()
    t.tokens = [b"hello", b" ", b"world", b"h", b"e", b"l", b"o", b"w", b"r", b"d"]
    t._index = {tok: i for i, tok in enumerate(t.tokens)}
    p = str(tmp_path / "vocab.bin")
    t.save(p)
    t2 = Tokenizer().load(p)
    assert t2.vocab_size == 10
    ids = t2.encode("hello world")
    assert t2.decode(ids) == "hello world"


def test_csv_loader(tmp_path):
    """Generic CSV loader (reference wifi_data_loader.hpp shape)."""
    import numpy as np
    from tnn_amd.data.loaders import CSVLoader
    p = tmp_path / "w.csv"
    p.write_text("a,b,c,y\n" + "\n".join(
        f"{i},{i*2},{i%3},{i*0.5}" for i in range(40)))
    dl = CSVLoader(str(p), batch_size=8)
    dl.load_data()
    assert dl.x.shape == (40, 3) and dl.y.shape == (40, 1)
    assert abs(dl.x.mean().item()) < 1e-5  # normalized
    xb, yb = next(iter(dl))
    assert xb.shape == (8, 3)
    # classification mode: int64 labels, unnormalized targets
    dl2 = CSVLoader(str(p), regression=False, target_cols=[2], batch_size=8)
    dl2.load_data()
    assert dl2.y.dtype.is_floating_point is False


def test_prepare_tokens_roundtrip(tmp_path):
    """tools/prepare_tokens.py builds a reference-format vocab.bin +
    uint16 train.bin that OpenWebTextLoader and Tokenizer can read."""
    import subprocess, sys, os
    import numpy as np
    from tnn_amd.data.tokenizer import Tokenizer
    from tnn_amd.data.loaders import OpenWebTextLoader
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    corpus = tmp_path / "corpus.txt"
    corpus.write_text("the quick brown fox jumps over the lazy dog " * 200)
    vocab = tmp_path / "vocab.bin"
    train = tmp_path / "train.bin"
    tool = os.path.join(root, "tools", "prepare_tokens.py")
    subprocess.run([sys.executable, tool, "build-vocab", str(corpus),
                    "--out", str(vocab), "--vocab-size", "300"], check=True)
    subprocess.run([sys.executable, tool, "tokenize", str(corpus),
                    "--vocab", str(vocab), "--out", str(train)], check=True)
    tok = Tokenizer().load(str(vocab))
    assert 256 < tok.vocab_size <= 300
    ids = np.fromfile(train, dtype=np.uint16)
    assert len(ids) > 100 and ids.max() < tok.vocab_size
    # decode returns the original text (byte-level BPE is lossless)
    text = tok.decode([int(i) for i in ids[:50]])
    assert "quick" in text or "fox" in text
    dl = OpenWebTextLoader(str(train), seq_len=32, samples_per_epoch=64,
                           batch_size=8)
    x, y = next(iter(dl))
    assert x.shape == (8, 32) and y.shape == (8, 32)
