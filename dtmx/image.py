"""Image API (reference python/mxnet/image/image.py subset): decode and the
augmenter primitives the data pipeline uses, exposed at the Python level.
The C++ loader (csrc/recordio.cpp) applies these natively in its worker
threads; this module is the scripting/debugging surface and the oracle the
augmentation tests compare against."""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def imdecode(buf: bytes, to_rgb: bool = True, flag: int = 1) -> torch.Tensor:
    """JPEG bytes -> uint8 HWC tensor (reference mx.image.imdecode; libjpeg
    via the native extension)."""
    from .ops.hip import require_ext

    ext = require_ext()
    return ext.decode_jpeg(buf, 3 if flag else 1)


def imresize(img: torch.Tensor, w: int, h: int) -> torch.Tensor:
    """Bilinear resize of a uint8 HWC image (reference mx.image.imresize)."""
    f = img.permute(2, 0, 1).unsqueeze(0).float()
    out = torch.nn.functional.interpolate(f, size=(h, w), mode="bilinear",
                                          align_corners=False)
    return out.squeeze(0).permute(1, 2, 0).round().clamp(0, 255).to(torch.uint8)


def resize_short(img: torch.Tensor, size: int) -> torch.Tensor:
    """Resize the shorter side to `size`, keeping aspect (reference
    mx.image.resize_short — the loader's `resize` augmenter)."""
    h, w = img.shape[0], img.shape[1]
    if h < w:
        return imresize(img, max(1, w * size // h), size)
    return imresize(img, size, max(1, h * size // w))


def center_crop(img: torch.Tensor, size: Tuple[int, int]):
    th, tw = size
    h, w = img.shape[0], img.shape[1]
    y0, x0 = (h - th) // 2, (w - tw) // 2
    return img[y0:y0 + th, x0:x0 + tw], (x0, y0, tw, th)


def random_crop(img: torch.Tensor, size: Tuple[int, int],
                rng: Optional[torch.Generator] = None):
    th, tw = size
    h, w = img.shape[0], img.shape[1]
    y0 = int(torch.randint(0, h - th + 1, (1,), generator=rng).item())
    x0 = int(torch.randint(0, w - tw + 1, (1,), generator=rng).item())
    return img[y0:y0 + th, x0:x0 + tw], (x0, y0, tw, th)


def horizontal_flip(img: torch.Tensor) -> torch.Tensor:
    return torch.flip(img, dims=[1])
