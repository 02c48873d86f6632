"""Multimodal (Imagen) datasets.

Reference: ppfleetx/data/dataset/multimodal_dataset.py (ImagenDataset:
base64 image filelists + tokenized captions). This environment has no
network/datasets, so the offline counterpart is a synthetic set with the
same sample contract: (image [3,H,W] in [-1,1], text_ids [L], text_mask
[L]) consumed by ImagenModule.training_step.
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset


class SyntheticImagenDataset(Dataset):
    collate_fn = None  # default collate: (imgs, ids, masks) batch tensors

    def __init__(self, num_samples: int = 1000, image_size: int = 64,
                 text_len: int = 32, text_vocab: int = 512, seed: int = 0,
                 **unused):
        self.num_samples = int(num_samples)
        self.image_size = int(image_size)
        self.text_len = int(text_len)
        self.text_vocab = int(text_vocab)
        self.seed = int(seed)

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + int(idx))
        img = torch.rand(3, self.image_size, self.image_size,
                         generator=g) * 2.0 - 1.0
        ids = torch.randint(1, self.text_vocab, (self.text_len,),
                            generator=g)
        nvalid = int(torch.randint(4, self.text_len + 1, (1,),
                                   generator=g))
        mask = torch.zeros(self.text_len, dtype=torch.long)
        mask[:nvalid] = 1
        ids = ids * mask
        return img, ids, mask
