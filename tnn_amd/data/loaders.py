"""Data loaders (reference include/data_loading/data_loader.hpp:25-118).

All image loaders emit NHWC float batches (the reference's new-layer-family
convention, cifar10_data_loader.hpp:37: stored CHW uint8 → NHWC float).
Synthetic loaders generate random data of the named dataset's shape — the
benchmark path (no network ⇒ no real datasets in this environment).
"""

from __future__ import annotations

import os
from typing import Iterator, Optional, Tuple

import numpy as np
import torch

from .augment import AugmentationStrategy


class BaseDataLoader:
    """Iterable over (x, y) batches; seedable; pluggable augmentation."""

    def __init__(self, batch_size: int = 128, shuffle: bool = True,
                 seed: int = 0, augmentation: Optional[AugmentationStrategy] = None,
                 dtype: torch.dtype = torch.float32):
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.rng = np.random.default_rng(seed)
        self.augmentation = augmentation
        self.dtype = dtype
        self.x: Optional[torch.Tensor] = None   # [N, ...] full dataset
        self.y: Optional[torch.Tensor] = None

    def load_data(self):
        raise NotImplementedError

    def _ensure(self):
        if self.x is None:
            self.load_data()

    def size(self) -> int:
        self._ensure()
        return self.x.shape[0]

    def get_data_shape(self) -> Tuple[int, ...]:
        self._ensure()
        return tuple(self.x.shape[1:])

    def __len__(self):
        return self.size() // self.batch_size

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        self._ensure()
        n = self.size()
        idx = np.arange(n)
        if self.shuffle:
            self.rng.shuffle(idx)
        for b in range(n // self.batch_size):
            sel = idx[b * self.batch_size:(b + 1) * self.batch_size]
            xb = self.x[sel].to(self.dtype)
            yb = self.y[sel]
            if self.augmentation is not None:
                xb = self.augmentation(xb, self.rng)
            yield xb, yb


class SyntheticImageLoader(BaseDataLoader):
    """Random images + labels of a given shape — the bench data source."""

    def __init__(self, shape=(32, 32, 3), num_classes=100, num_samples=2048,
                 **kw):
        super().__init__(**kw)
        self.shape, self.num_classes, self.num_samples = shape, num_classes, num_samples

    def load_data(self):
        g = torch.Generator().manual_seed(int(self.rng.integers(2 ** 31)))
        self.x = torch.randn(self.num_samples, *self.shape, generator=g)
        self.y = torch.randint(0, self.num_classes, (self.num_samples,), generator=g)


class SyntheticTokenLoader(BaseDataLoader):
    def __init__(self, seq_len=1024, vocab_size=50257, num_samples=512, **kw):
        super().__init__(**kw)
        self.seq_len, self.vocab_size, self.num_samples = seq_len, vocab_size, num_samples

    def load_data(self):
        g = torch.Generator().manual_seed(int(self.rng.integers(2 ** 31)))
        tokens = torch.randint(0, self.vocab_size, (self.num_samples, self.seq_len + 1),
                               generator=g)
        self.x, self.y = tokens[:, :-1], tokens[:, 1:]


class RegressionLoader(BaseDataLoader):
    def __init__(self, in_dim=16, out_dim=1, num_samples=4096, **kw):
        super().__init__(**kw)
        self.in_dim, self.out_dim, self.num_samples = in_dim, out_dim, num_samples

    def load_data(self):
        g = torch.Generator().manual_seed(int(self.rng.integers(2 ** 31)))
        self.x = torch.randn(self.num_samples, self.in_dim, generator=g)
        w = torch.randn(self.in_dim, self.out_dim, generator=g)
        self.y = self.x @ w + 0.01 * torch.randn(self.num_samples, self.out_dim,
                                                 generator=g)


class MNISTLoader(BaseDataLoader):
    """MNIST from the CSV format the reference uses
    (reference include/data_loading/mnist_data_loader.hpp)."""

    def __init__(self, path: str, train: bool = True, **kw):
        super().__init__(**kw)
        self.path, self.train = path, train

    def load_data(self):
        fname = os.path.join(self.path, "mnist_train.csv" if self.train
                             else "mnist_test.csv")
        raw = np.loadtxt(fname, delimiter=",", skiprows=1)
        self.y = torch.from_numpy(raw[:, 0].astype(np.int64))
        x = raw[:, 1:].astype(np.float32).reshape(-1, 28, 28, 1) / 255.0
        self.x = torch.from_numpy(x)


class CIFAR10Loader(BaseDataLoader):
    """CIFAR-10 binary batches, stored CHW uint8 → NHWC float
    (reference cifar10_data_loader.hpp:37,151)."""

    files_train = [f"data_batch_{i}.bin" for i in range(1, 6)]
    files_test = ["test_batch.bin"]
    label_bytes = 1
    num_classes = 10

    def __init__(self, path: str, train: bool = True, **kw):
        super().__init__(**kw)
        self.path, self.train = path, train

    def load_data(self):
        files = self.files_train if self.train else self.files_test
        xs, ys = [], []
        rec = self.label_bytes + 3072
        for fn in files:
            raw = np.fromfile(os.path.join(self.path, fn), dtype=np.uint8)
            raw = raw.reshape(-1, rec)
            ys.append(raw[:, self.label_bytes - 1].astype(np.int64))
            img = raw[:, self.label_bytes:].reshape(-1, 3, 32, 32)
            xs.append(np.transpose(img, (0, 2, 3, 1)))  # CHW -> HWC
        self.y = torch.from_numpy(np.concatenate(ys))
        self.x = torch.from_numpy(
            np.concatenate(xs).astype(np.float32) / 255.0)


class CIFAR100Loader(CIFAR10Loader):
    """CIFAR-100: 2 label bytes (coarse, fine); fine label used."""

    files_train = ["train.bin"]
    files_test = ["test.bin"]
    label_bytes = 2
    num_classes = 100


class TinyImageNetLoader(BaseDataLoader):
    """Tiny-ImageNet, loaded either from the RAW dataset layout
    (``<path>/<split>/<class>/**/*.JPEG``, decoded by the in-tree C++
    codec — reference src/data_loading/stb_image_impl.cpp analog) or
    from pre-decoded ``{split}_{x,y}.npy`` when present (fast path)."""

    image_size = 64

    def __init__(self, path: str, train: bool = True, **kw):
        super().__init__(**kw)
        self.path, self.train = path, train

    def load_data(self):
        split = "train" if self.train else "val"
        npx = os.path.join(self.path, f"{split}_x.npy")
        if os.path.exists(npx):
            self.x = torch.from_numpy(
                np.load(npx).astype(np.float32) / 255.0)
            self.y = torch.from_numpy(
                np.load(os.path.join(self.path, f"{split}_y.npy"))
                .astype(np.int64))
            return
        from .imageio import load_image_dir
        x, self.y = load_image_dir(os.path.join(self.path, split),
                                   size=self.image_size)
        self.x = x.float() / 255.0


class ImageNet100Loader(TinyImageNetLoader):
    """ImageNet-100: same dual raw-directory / .npy loading as
    Tiny-ImageNet, 224x224 center-fit (reference decodes via stb_image)."""

    image_size = 224


class CSVLoader(BaseDataLoader):
    """Generic CSV feature/target loader with column ranges, optional
    header skip and z-score normalization (reference
    include/data_loading/wifi_data_loader.hpp — the WiFi RSSI dataset is
    one instance of this shape)."""

    def __init__(self, path: str, feature_cols=None, target_cols=None,
                 has_header: bool = True, regression: bool = True,
                 normalize: bool = True, **kw):
        super().__init__(**kw)
        self.path = path
        self.feature_cols, self.target_cols = feature_cols, target_cols
        self.has_header, self.regression = has_header, regression
        self.normalize = normalize

    def load_data(self):
        rows = []
        with open(self.path) as f:
            for i, line in enumerate(f):
                if i == 0 and self.has_header:
                    continue
                cells = line.strip().split(",")
                if cells and any(cells):
                    rows.append([float(c) for c in cells if c != ""])
        if not rows:
            raise ValueError(f"no rows in {self.path}")
        arr = np.asarray(rows, dtype=np.float32)
        ncol = arr.shape[1]
        fcols = (list(self.feature_cols) if self.feature_cols is not None
                 else list(range(ncol - 1)))
        tcols = (list(self.target_cols) if self.target_cols is not None
                 else [ncol - 1])
        x = arr[:, fcols]
        t = arr[:, tcols]
        if self.normalize:
            mu, sd = x.mean(0), x.std(0)
            sd[sd == 0] = 1.0
            x = (x - mu) / sd
            self.feature_stats = (mu, sd)
            if self.regression:
                tmu, tsd = t.mean(0), t.std(0)
                tsd[tsd == 0] = 1.0
                t = (t - tmu) / tsd
                self.target_stats = (tmu, tsd)
        self.x = torch.from_numpy(x)
        self.y = (torch.from_numpy(t) if self.regression
                  else torch.from_numpy(t[:, 0].astype(np.int64)))


class OpenWebTextLoader(BaseDataLoader):
    """mmap'd uint16 GPT-2 token file, random windows
    (reference include/data_loading/open_webtext_data_loader.hpp:11-95).
    Emits int64 ids + next-token targets (class ids, not one-hot — the
    fused CE loss takes ids directly)."""

    def __init__(self, bin_path: str, seq_len: int = 1024, samples_per_epoch=4096,
                 **kw):
        super().__init__(**kw)
        self.bin_path, self.seq_len = bin_path, seq_len
        self.samples_per_epoch = samples_per_epoch
        self._mm = None

    def load_data(self):
        self._mm = np.memmap(self.bin_path, dtype=np.uint16, mode="r")
        self.x = torch.empty(0)  # windows are sampled lazily
        self.y = torch.empty(0)

    def size(self):
        self._ensure()
        return self.samples_per_epoch

    def get_data_shape(self):
        return (self.seq_len,)

    def __iter__(self):
        self._ensure()
        n_tok = len(self._mm) - self.seq_len - 1
        for _ in range(self.samples_per_epoch // self.batch_size):
            starts = self.rng.integers(0, n_tok, self.batch_size)
            xb = np.stack([self._mm[s:s + self.seq_len] for s in starts])
            yb = np.stack([self._mm[s + 1:s + self.seq_len + 1] for s in starts])
            yield (torch.from_numpy(xb.astype(np.int64)),
                   torch.from_numpy(yb.astype(np.int64)))


class DataLoaderFactory:
    """reference include/data_loading/data_loader_factory.hpp:18."""

    _REGISTRY = {
        "synthetic_image": SyntheticImageLoader,
        "synthetic_tokens": SyntheticTokenLoader,
        "regression": RegressionLoader,
        "mnist": MNISTLoader,
        "cifar10": CIFAR10Loader,
        "cifar100": CIFAR100Loader,
        "tiny_imagenet": TinyImageNetLoader,
        "imagenet_100": ImageNet100Loader,
        "csv": CSVLoader,
        "wifi": CSVLoader,
        "openwebtext": OpenWebTextLoader,
    }

    @classmethod
    def create(cls, name: str, **kw) -> BaseDataLoader:
        return cls._REGISTRY[name](**kw)

    @classmethod
    def register(cls, name: str, loader_cls):
        cls._REGISTRY[name] = loader_cls
