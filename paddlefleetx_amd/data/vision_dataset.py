"""Vision datasets + train transforms.

Reference: ppfleetx/data/dataset/vision_dataset.py (ImageFolder ImageNet
:~40, CIFAR) and ppfleetx/data/transforms/preprocess.py (mixup/cutmix
:~300). The filesystem ImageNet reader keeps the same folder layout
(`root/class_x/*.jpg`); a synthetic variant generates fixed random tensors
for benchmarks (this environment has no dataset downloads).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset


class SyntheticImageNetDataset(Dataset):
    """Random images/labels of ImageNet shape, deterministic per index."""

    collate_fn = None  # default collate

    def __init__(self, num_samples: int = 10000, image_size: int = 224,
                 num_classes: int = 1000, mode: str = "Train",
                 moco_two_crops: bool = False, **unused):
        self.num_samples = int(num_samples)
        self.image_size = int(image_size)
        self.num_classes = int(num_classes)
        # MoCo contract: sample = ((crop_q, crop_k), label)
        self.moco_two_crops = bool(moco_two_crops)

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(int(idx))
        img = torch.randn(3, self.image_size, self.image_size, generator=g)
        label = int(torch.randint(0, self.num_classes, (1,), generator=g))
        if self.moco_two_crops:
            crop_k = img + 0.1 * torch.randn(img.shape, generator=g)
            return (img, crop_k), label
        return img, label


class ImageFolderDataset(Dataset):
    """`root/<class>/<img>` folder tree -> (CHW float tensor, class index).

    Decoding uses torchvision when present; else raw numpy for .npy files.
    """

    collate_fn = None

    def __init__(self, root: str, image_size: int = 224, mode: str = "Train",
                 rand_augment: bool = False, ra_num_ops: int = 2,
                 ra_magnitude: int = 9, transforms=None,
                 transform_ops=None, **unused):
        self.root = root
        self.image_size = image_size
        self.rand_augment = None
        if rand_augment and mode == "Train":
            self.rand_augment = RandAugment(ra_num_ops, ra_magnitude)
        self.transforms = None
        transforms = transforms or transform_ops  # reference key name
        if transforms:
            from paddlefleetx_amd.data.transforms import build_transforms
            self.transforms = build_transforms(transforms)
        classes = sorted(d for d in os.listdir(root)
                         if os.path.isdir(os.path.join(root, d)))
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = []
        for c in classes:
            cdir = os.path.join(root, c)
            for f in sorted(os.listdir(cdir)):
                self.samples.append((os.path.join(cdir, f),
                                     self.class_to_idx[c]))

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        path, label = self.samples[idx]
        if path.endswith(".npy"):
            arr = np.load(path)
            img = torch.from_numpy(arr).float()
        else:
            try:
                from PIL import Image
                im = Image.open(path).convert("RGB").resize(
                    (self.image_size, self.image_size))
                img = torch.from_numpy(np.asarray(im)).permute(2, 0, 1) \
                    .float() / 255.0
            except ImportError as e:
                raise RuntimeError(f"cannot decode {path}: PIL missing") from e
        if self.transforms is not None:
            img = self.transforms(img)
        if self.rand_augment is not None:
            img = self.rand_augment(img)
        return img, label


# ---------------------------------------------------------------------------
# batch transforms (reference data/transforms/preprocess.py mixup/cutmix)
# ---------------------------------------------------------------------------

def one_hot(labels: torch.Tensor, num_classes: int,
            smoothing: float = 0.0) -> torch.Tensor:
    off = smoothing / num_classes
    on = 1.0 - smoothing + off
    out = torch.full((labels.shape[0], num_classes), off,
                     device=labels.device)
    out.scatter_(1, labels.unsqueeze(1), on)
    return out


def mixup_batch(images: torch.Tensor, labels: torch.Tensor,
                num_classes: int, alpha: float = 0.2,
                smoothing: float = 0.0
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    lam = float(np.random.beta(alpha, alpha)) if alpha > 0 else 1.0
    perm = torch.randperm(images.shape[0], device=images.device)
    mixed = lam * images + (1.0 - lam) * images[perm]
    y = one_hot(labels, num_classes, smoothing)
    y = lam * y + (1.0 - lam) * y[perm]
    return mixed, y


def cutmix_batch(images: torch.Tensor, labels: torch.Tensor,
                 num_classes: int, alpha: float = 1.0,
                 smoothing: float = 0.0
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    lam = float(np.random.beta(alpha, alpha)) if alpha > 0 else 1.0
    B, _, H, W = images.shape
    cut = (1.0 - lam) ** 0.5
    ch, cw = int(H * cut), int(W * cut)
    cy = np.random.randint(H)
    cx = np.random.randint(W)
    y1, y2 = max(cy - ch // 2, 0), min(cy + ch // 2, H)
    x1, x2 = max(cx - cw // 2, 0), min(cx + cw // 2, W)
    perm = torch.randperm(B, device=images.device)
    out = images.clone()
    out[:, :, y1:y2, x1:x2] = images[perm][:, :, y1:y2, x1:x2]
    lam_adj = 1.0 - ((y2 - y1) * (x2 - x1) / (H * W))
    y = one_hot(labels, num_classes, smoothing)
    y = lam_adj * y + (1.0 - lam_adj) * y[perm]
    return out, y


# ---------------------------------------------------------------------------
# RandAugment (reference data/transforms/preprocess.py rand-aug), done
# natively on CHW float tensors in [0, 1] — no PIL dependency.
# ---------------------------------------------------------------------------

def _affine(img: torch.Tensor, theta_2x3: torch.Tensor) -> torch.Tensor:
    c, h, w = img.shape
    grid = torch.nn.functional.affine_grid(
        theta_2x3.unsqueeze(0), (1, c, h, w), align_corners=False)
    return torch.nn.functional.grid_sample(
        img.unsqueeze(0), grid, padding_mode="zeros",
        align_corners=False).squeeze(0)


class RandAugment:
    """Pick `num_ops` random ops at strength `magnitude`/`num_bins` per
    image. Ops: brightness, contrast, sharpness, solarize, posterize,
    autocontrast, rotate, shear-x/y, translate-x/y (the affine ones via
    grid_sample). Callable on a CHW float tensor in [0, 1]."""

    OPS = ("identity", "brightness", "contrast", "sharpness", "solarize",
           "posterize", "autocontrast", "rotate", "shear_x", "shear_y",
           "translate_x", "translate_y")

    def __init__(self, num_ops: int = 2, magnitude: int = 9,
                 num_bins: int = 31, generator: "torch.Generator" = None):
        self.num_ops = int(num_ops)
        self.m = float(magnitude) / float(num_bins - 1)
        self.gen = generator

    def _rand(self, n):
        return torch.rand(n, generator=self.gen)

    def _apply(self, img: torch.Tensor, op: str, sign: float) -> torch.Tensor:
        m = self.m
        if op == "identity":
            return img
        if op == "brightness":
            return (img * (1.0 + sign * 0.9 * m)).clamp(0, 1)
        if op == "contrast":
            mean = img.mean()
            return (mean + (img - mean) * (1.0 + sign * 0.9 * m)).clamp(0, 1)
        if op == "sharpness":
            k = torch.full((img.shape[0], 1, 3, 3), 1.0 / 9.0)
            blur = torch.nn.functional.conv2d(
                img.unsqueeze(0), k, padding=1,
                groups=img.shape[0]).squeeze(0)
            return (img + sign * 0.9 * m * (img - blur)).clamp(0, 1)
        if op == "solarize":
            thr = 1.0 - m  # magnitude lowers the inversion threshold
            return torch.where(img >= thr, 1.0 - img, img)
        if op == "posterize":
            bits = max(1, 8 - int(round(4 * m)))
            q = float(1 << (8 - bits))
            return torch.floor(img * 255.0 / q) * q / 255.0
        if op == "autocontrast":
            lo = img.amin(dim=(1, 2), keepdim=True)
            hi = img.amax(dim=(1, 2), keepdim=True)
            scale = torch.where(hi > lo, 1.0 / (hi - lo).clamp_min(1e-6),
                                torch.ones_like(hi))
            return ((img - lo) * scale).clamp(0, 1)
        # affine ops
        t = torch.eye(2, 3)
        if op == "rotate":
            a = sign * m * (30.0 * 3.141592653589793 / 180.0)
            ca, sa = float(torch.cos(torch.tensor(a))), \
                float(torch.sin(torch.tensor(a)))
            t[0, 0], t[0, 1], t[1, 0], t[1, 1] = ca, -sa, sa, ca
        elif op == "shear_x":
            t[0, 1] = sign * 0.3 * m
        elif op == "shear_y":
            t[1, 0] = sign * 0.3 * m
        elif op == "translate_x":
            t[0, 2] = sign * 0.45 * m
        elif op == "translate_y":
            t[1, 2] = sign * 0.45 * m
        return _affine(img, t)

    def __call__(self, img: torch.Tensor) -> torch.Tensor:
        for _ in range(self.num_ops):
            op = self.OPS[int(self._rand(1) * len(self.OPS))]
            sign = 1.0 if float(self._rand(1)) < 0.5 else -1.0
            img = self._apply(img, op, sign)
        return img


class ContrativeLearningDataset(ImageFolderDataset):
    """Two-view contrastive dataset (reference vision_dataset.py:379,
    keeping its spelling): each sample is ((view_q, view_k), label) where
    both views are independent draws of the transform pipeline."""

    def __getitem__(self, idx):
        path, label = self.samples[idx]
        base = super().__getitem__(idx)[0] if self.transforms is None \
            else None
        if self.transforms is None:
            # no pipeline configured: jitter the second view slightly
            return (base, base + 0.05 * torch.randn_like(base)), label
        if path.endswith(".npy"):
            img = torch.from_numpy(np.load(path)).float()
        else:
            from PIL import Image
            im = Image.open(path).convert("RGB").resize(
                (self.image_size, self.image_size))
            img = torch.from_numpy(np.asarray(im)).permute(2, 0, 1) \
                .float() / 255.0
        return (self.transforms(img), self.transforms(img)), label


class CIFAR10Dataset(Dataset):
    """CIFAR-10 from the standard python-pickle batches on local disk
    (reference vision_dataset.py:302; no download here — point `root` at
    an extracted cifar-10-batches-py directory)."""

    collate_fn = None

    def __init__(self, root: str, mode: str = "Train", transforms=None,
                 transform_ops=None, **unused):
        import pickle
        names = [f"data_batch_{i}" for i in range(1, 6)] \
            if mode == "Train" else ["test_batch"]
        datas, labels = [], []
        for n in names:
            path = os.path.join(root, n)
            if not os.path.exists(path):
                continue
            with open(path, "rb") as f:
                d = pickle.load(f, encoding="bytes")
            datas.append(np.asarray(d[b"data"] if b"data" in d
                                    else d["data"]))
            labels.extend(d[b"labels"] if b"labels" in d else d["labels"])
        assert datas, f"no CIFAR-10 batches under {root}"
        self.images = np.concatenate(datas).reshape(-1, 3, 32, 32)
        self.labels = np.asarray(labels, dtype=np.int64)
        self.transforms = None
        ops = transforms or transform_ops
        if ops:
            from paddlefleetx_amd.data.transforms import build_transforms
            self.transforms = build_transforms(ops)

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        img = torch.from_numpy(self.images[idx].astype(np.float32) / 255.0)
        if self.transforms is not None:
            img = self.transforms(img)
        return img, int(self.labels[idx])
