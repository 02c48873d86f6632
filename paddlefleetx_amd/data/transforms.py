"""Vision transform ops (reference ppfleetx/data/transforms/preprocess.py:
DecodeImage:40, ResizeImage:107, CenterCropImage:142, RandCropImage:162,
RandFlipImage:211, NormalizeImage:231, ToCHWImage:280, ColorJitter:294,
GaussianBlur:313) re-expressed natively on torch tensors — no PIL/cv2
dependency; images are CHW float tensors in [0, 1] (HWC uint8 inputs are
converted by ToCHWImage / DecodeImage).

`build_transforms([...])` mirrors the reference's config-driven op list:
each entry is `{OpName: {kwargs}}` exactly as the YAML Data.transforms
section writes them.
"""

from __future__ import annotations

import math
from typing import List, Optional, Sequence

import numpy as np
import torch
import torch.nn.functional as F

__all__ = ["DecodeImage", "ResizeImage", "CenterCropImage", "RandCropImage",
           "RandFlipImage", "NormalizeImage", "ToCHWImage", "ColorJitter",
           "GaussianBlur", "Compose", "build_transforms"]


def _chw(img: torch.Tensor) -> torch.Tensor:
    if img.dim() == 3 and img.shape[0] not in (1, 3) and img.shape[-1] in (1, 3):
        img = img.permute(2, 0, 1)
    return img.float()


class DecodeImage:
    """ndarray/uint8 -> CHW float in [0,1] (reference :40 decodes jpeg
    bytes; here the dataset already hands us arrays)."""

    def __init__(self, to_rgb: bool = True, channel_first: bool = False):
        self.channel_first = channel_first

    def __call__(self, img):
        if isinstance(img, np.ndarray):
            img = torch.from_numpy(np.ascontiguousarray(img))
        if img.dtype == torch.uint8:
            img = img.float() / 255.0
        if not self.channel_first:
            img = _chw(img)
        return img


class ResizeImage:
    def __init__(self, size=None, resize_short=None, interpolation="bilinear",
                 **unused):
        assert size is not None or resize_short is not None
        self.size = (size, size) if isinstance(size, int) else size
        self.resize_short = resize_short
        self.mode = interpolation if interpolation in ("bilinear", "nearest",
                                                       "bicubic") else "bilinear"

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        _, h, w = img.shape
        if self.resize_short is not None:
            scale = self.resize_short / min(h, w)
            out = (int(round(h * scale)), int(round(w * scale)))
        else:
            out = (self.size[1], self.size[0]) if self.size else (h, w)
        kw = {} if self.mode == "nearest" else {"align_corners": False}
        return F.interpolate(img.unsqueeze(0), size=out, mode=self.mode,
                             **kw).squeeze(0)


class CenterCropImage:
    def __init__(self, size, **unused):
        self.size = (size, size) if isinstance(size, int) else size

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        _, h, w = img.shape
        th, tw = self.size
        y = max(0, (h - th) // 2)
        x = max(0, (w - tw) // 2)
        return img[:, y:y + th, x:x + tw]


class RandCropImage:
    """Random resized crop (reference :162: random scale/ratio window,
    resized to `size`)."""

    def __init__(self, size, scale=(0.08, 1.0), ratio=(3. / 4, 4. / 3),
                 interpolation="bilinear", generator=None, **unused):
        self.size = (size, size) if isinstance(size, int) else size
        self.scale, self.ratio = scale, ratio
        self.resize = ResizeImage(size=self.size[0],
                                  interpolation=interpolation)
        self.gen = generator

    def _rand(self):
        return float(torch.rand((), generator=self.gen))

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        _, h, w = img.shape
        area = h * w
        for _ in range(10):
            target = area * (self.scale[0] +
                             (self.scale[1] - self.scale[0]) * self._rand())
            logr = (math.log(self.ratio[0]) +
                    (math.log(self.ratio[1]) - math.log(self.ratio[0]))
                    * self._rand())
            ar = math.exp(logr)
            cw = int(round(math.sqrt(target * ar)))
            ch = int(round(math.sqrt(target / ar)))
            if cw <= w and ch <= h and cw > 0 and ch > 0:
                y = int(self._rand() * (h - ch + 1))
                x = int(self._rand() * (w - cw + 1))
                crop = img[:, y:y + ch, x:x + cw]
                return self.resize(crop)
        return self.resize(CenterCropImage(min(h, w))(img))


class RandFlipImage:
    def __init__(self, flip_code: int = 1, generator=None, **unused):
        self.flip_code = flip_code  # 1: horizontal (reference :219)
        self.gen = generator

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        if float(torch.rand((), generator=self.gen)) < 0.5:
            dim = 2 if self.flip_code == 1 else 1
            img = torch.flip(img, dims=(dim,))
        return img


class NormalizeImage:
    def __init__(self, scale=None, mean=None, std=None, order="chw",
                 **unused):
        if isinstance(scale, str):
            # YAML writes "1./255." — parse the fraction without eval
            num, _, den = scale.partition("/")
            scale = float(num) / float(den) if den else float(num)
        self.scale = scale if scale is not None else 1.0
        mean = mean if mean is not None else [0.485, 0.456, 0.406]
        std = std if std is not None else [0.229, 0.224, 0.225]
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        if img.max() > 1.5 and self.scale != 1.0:
            img = img * self.scale  # uint8-range input: bring to [0,1]
        return (img - self.mean) / self.std


class ToCHWImage:
    def __call__(self, img):
        if isinstance(img, np.ndarray):
            img = torch.from_numpy(np.ascontiguousarray(img))
        return _chw(img)


class ColorJitter:
    """brightness/contrast/saturation/hue jitter (reference :294); hue
    via a luma-preserving channel rotation approximation."""

    def __init__(self, brightness=0.0, contrast=0.0, saturation=0.0,
                 hue=0.0, p: float = 1.0, generator=None, **unused):
        self.b, self.c, self.s, self.h = brightness, contrast, saturation, hue
        self.p = float(p)  # RandomApply probability (reference moco v2)
        self.gen = generator

    def _f(self, mag):
        r = float(torch.rand((), generator=self.gen)) * 2 - 1
        return 1.0 + r * mag

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        if self.p < 1.0 and float(torch.rand((), generator=self.gen)) >= self.p:
            return img
        if self.b:
            img = (img * self._f(self.b)).clamp(0, 1)
        if self.c:
            mean = img.mean()
            img = (mean + (img - mean) * self._f(self.c)).clamp(0, 1)
        if self.s and img.shape[0] == 3:
            gray = (0.299 * img[0] + 0.587 * img[1] + 0.114 * img[2]) \
                .unsqueeze(0)
            img = (gray + (img - gray) * self._f(self.s)).clamp(0, 1)
        if self.h and img.shape[0] == 3:
            shift = (self._f(self.h) - 1.0)  # in [-h, h]
            r, g, b = img[0], img[1], img[2]
            img = torch.stack([
                (r + shift * (g - r)).clamp(0, 1),
                (g + shift * (b - g)).clamp(0, 1),
                (b + shift * (r - b)).clamp(0, 1)])
        return img


class GaussianBlur:
    def __init__(self, sigma=(0.1, 2.0), kernel_size: int = 9,
                 p: float = 1.0, generator=None, **unused):
        self.sigma = sigma if isinstance(sigma, (tuple, list)) \
            else (sigma, sigma)
        self.k = kernel_size | 1  # odd
        self.p = float(p)
        self.gen = generator

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        img = _chw(img)
        if self.p < 1.0 and float(torch.rand((), generator=self.gen)) >= self.p:
            return img
        lo, hi = self.sigma
        s = lo + (hi - lo) * float(torch.rand((), generator=self.gen))
        half = self.k // 2
        xs = torch.arange(-half, half + 1, dtype=torch.float32)
        g = torch.exp(-(xs ** 2) / (2 * s * s))
        g = (g / g.sum()).view(1, 1, -1)
        c = img.shape[0]
        out = img.unsqueeze(0)
        out = F.conv2d(out, g.view(1, 1, 1, -1).expand(c, 1, 1, self.k),
                       padding=(0, half), groups=c)
        out = F.conv2d(out, g.view(1, 1, -1, 1).expand(c, 1, self.k, 1),
                       padding=(half, 0), groups=c)
        return out.squeeze(0)


class Compose:
    def __init__(self, ops: Sequence):
        self.ops = list(ops)

    def __call__(self, img):
        for op in self.ops:
            img = op(img)
        return img


def build_transforms(cfg_list: Optional[List[dict]],
                     generator=None) -> Optional[Compose]:
    """Config-driven assembly (reference transform op lists in the vis
    YAMLs): [{OpName: {kwargs}}, ...] -> Compose. Unknown names raise."""
    if not cfg_list:
        return None
    table = {c.__name__: c for c in
             (DecodeImage, ResizeImage, CenterCropImage, RandCropImage,
              RandFlipImage, NormalizeImage, ToCHWImage, ColorJitter,
              GaussianBlur)}
    from paddlefleetx_amd.data.vision_dataset import RandAugment
    table["RandAugment"] = RandAugment
    ops = []
    for entry in cfg_list:
        (name, kw), = entry.items() if isinstance(entry, dict) \
            else ((entry, {}),)
        kw = dict(kw or {})
        if name not in table:
            raise ValueError(f"unknown transform {name}")
        cls = table[name]
        try:
            ops.append(cls(generator=generator, **kw))
        except TypeError:
            ops.append(cls(**kw))
    return Compose(ops)
