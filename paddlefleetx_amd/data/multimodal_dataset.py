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


def get_keys(data_path: str, gpu_num: int, rank: int = None):
    """Rank-sharded filelist selection (reference multimodal_dataset.py:
    40-60): pad the shard list to a multiple of gpu_num, then stride by
    rank so every rank gets an equal number of shard files."""
    import random
    files = [line.strip() for line in open(data_path)
             if line.strip() != ""]
    if rank is None:
        from paddlefleetx_amd.parallel.env import get_hcg
        try:
            rank = get_hcg().global_rank
        except Exception:
            rank = 0
    if files and len(files) % gpu_num != 0:
        added = gpu_num - (len(files) % gpu_num)
        files = files + [random.choice(files) for _ in range(added)]
    return files[rank::gpu_num]


class ImagenFileDataset(Dataset):
    """Filelist-of-shards Imagen dataset (reference ImagenDataset:62 —
    base64 tsv shards). Offline form: each shard file holds lines
    `image_path<TAB>caption`; images are .npy CHW arrays (jpeg decode
    would need PIL, absent here). Captions are tokenized when a
    tokenizer is given, else hashed to synthetic ids of `text_len`."""

    collate_fn = None

    def __init__(self, input_path: str, image_size: int = 64,
                 text_max_len: int = 128, tokenizer=None, gpu_num: int = 1,
                 rank: int = None, mode: str = "Train", **unused):
        import os
        self.rows = []
        base = os.path.dirname(os.path.abspath(input_path))
        for shard in get_keys(input_path, gpu_num, rank):
            sp = shard if os.path.isabs(shard) else os.path.join(base, shard)
            for line in open(sp):
                if "\t" in line:
                    img, cap = line.rstrip("\n").split("\t", 1)
                    self.rows.append((
                        img if os.path.isabs(img) else os.path.join(
                            os.path.dirname(sp), img), cap))
        self.image_size = int(image_size)
        self.text_max_len = int(text_max_len)
        self.tokenizer = tokenizer

    def __len__(self):
        return len(self.rows)

    def __getitem__(self, idx):
        import numpy as np
        path, cap = self.rows[idx]
        img = torch.from_numpy(np.load(path)).float()
        if img.max() > 1.5:
            img = img / 255.0
        img = img * 2.0 - 1.0  # [-1, 1] (reference normalization)
        L = self.text_max_len
        if self.tokenizer is not None:
            ids = self.tokenizer.encode(cap)[:L]
        else:
            ids = [1 + (hash(w) % 510) for w in cap.split()[:L]]
        mask = torch.zeros(L, dtype=torch.long)
        mask[:len(ids)] = 1
        out = torch.zeros(L, dtype=torch.long)
        out[:len(ids)] = torch.tensor(ids, dtype=torch.long)
        return img, out, mask
