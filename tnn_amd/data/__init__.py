"""Data loading + augmentation (reference include/data_loading/,
include/data_augmentation/)."""

from .loaders import (BaseDataLoader, SyntheticImageLoader, SyntheticTokenLoader,
                      MNISTLoader, CIFAR10Loader, CIFAR100Loader,
                      TinyImageNetLoader, ImageNet100Loader, OpenWebTextLoader, RegressionLoader,
                      DataLoaderFactory)
from .augment import (Augmentation, HorizontalFlip, RandomCrop, Rotate,
                      Brightness, Contrast, GaussianNoise, Cutout, Normalize,
                      AugmentationStrategy)
from .tokenizer import Tokenizer

__all__ = [
    "BaseDataLoader", "SyntheticImageLoader", "SyntheticTokenLoader",
    "MNISTLoader", "CIFAR10Loader", "CIFAR100Loader", "TinyImageNetLoader", "ImageNet100Loader",
    "OpenWebTextLoader", "RegressionLoader", "DataLoaderFactory",
    "Augmentation", "HorizontalFlip", "RandomCrop", "Rotate", "Brightness",
    "Contrast", "GaussianNoise", "Cutout", "Normalize", "AugmentationStrategy",
    "Tokenizer",
]
