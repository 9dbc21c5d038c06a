"""In-tree image decoding (reference src/data_loading/stb_image_impl.cpp
analog): baseline JPEG / PNG / BMP decoded by the from-scratch C++ codec
in csrc/imagecodec.cpp — raw Tiny-ImageNet / ImageNet-style datasets load
without an offline preprocessing step."""

from __future__ import annotations

import os

import torch

from .. import _C


def decode_image(data: bytes) -> torch.Tensor:
    """Decode an encoded image to a uint8 [H, W, C] tensor (C = 1..4)."""
    return _C.ext_any().decode_image(data)


def load_image(path: str, channels: int = 3) -> torch.Tensor:
    """Read + decode; convert to ``channels`` (1 or 3) like stbi_load's
    desired_channels."""
    with open(path, "rb") as f:
        img = decode_image(f.read())
    c = img.shape[-1]
    if channels == 3:
        if c == 1:
            img = img.repeat(1, 1, 3)
        elif c == 2:  # gray+alpha: drop alpha, replicate gray
            img = img[..., :1].repeat(1, 1, 3)
        elif c == 4:  # RGBA: drop alpha
            img = img[..., :3].contiguous()
    elif channels == 1:
        if c >= 3:
            f32 = img[..., :3].float()
            img = (0.299 * f32[..., 0] + 0.587 * f32[..., 1]
                   + 0.114 * f32[..., 2]).round().clamp(0, 255).to(
                       torch.uint8).unsqueeze(-1)
        elif c == 2:
            img = img[..., :1].contiguous()
    return img


IMAGE_EXTS = (".jpg", ".jpeg", ".png", ".bmp", ".JPEG", ".JPG", ".PNG")


def is_image_file(name: str) -> bool:
    return name.endswith(IMAGE_EXTS)


def load_image_dir(root: str, size: int | None = None):
    """Decode every image under class subdirectories of ``root`` into
    (x [N,H,W,3] uint8, y [N]) with labels = sorted subdir index. Images
    must share one size unless ``size`` crops/pads to size x size."""
    classes = sorted(d for d in os.listdir(root)
                     if os.path.isdir(os.path.join(root, d)))
    xs, ys = [], []
    for label, cls in enumerate(classes):
        cdir = os.path.join(root, cls)
        for dirpath, _, files in os.walk(cdir):
            for fn in sorted(files):
                if not is_image_file(fn):
                    continue
                img = load_image(os.path.join(dirpath, fn), channels=3)
                if size is not None:
                    img = _center_fit(img, size)
                xs.append(img)
                ys.append(label)
    if not xs:
        raise FileNotFoundError(f"no images under {root}")
    return torch.stack(xs), torch.tensor(ys, dtype=torch.int64)


def _center_fit(img: torch.Tensor, size: int) -> torch.Tensor:
    """Center-crop (or zero-pad) to size x size."""
    h, w, c = img.shape
    out = torch.zeros(size, size, c, dtype=img.dtype)
    sh, sw = max(0, (h - size) // 2), max(0, (w - size) // 2)
    dh, dw = max(0, (size - h) // 2), max(0, (size - w) // 2)
    ch, cw = min(h, size), min(w, size)
    out[dh:dh + ch, dw:dw + cw] = img[sh:sh + ch, sw:sw + cw]
    return out
