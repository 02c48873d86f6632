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
                 **unused):
        self.root = root
        self.image_size = image_size
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
