"""build_dataloader (reference ppfleetx/data/__init__.py:28-119)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch
from torch.utils.data import DataLoader

from paddlefleetx_amd.data.gpt_dataset import (BlendedGPTDataset, GPTDataset,
                                               GPTSyntheticDataset)
from paddlefleetx_amd.data.sampler import GPTBatchSampler
from paddlefleetx_amd.data.ernie_dataset import (ErnieSyntheticDataset,
                                                 ErnieWWMDataset)
from paddlefleetx_amd.data.vision_dataset import (
    CIFAR10Dataset, ContrativeLearningDataset, ImageFolderDataset,
    SyntheticImageNetDataset)
from paddlefleetx_amd.parallel.env import (get_data_world_rank,
                                           get_data_world_size)
from paddlefleetx_amd.utils.log import logger

_DATASETS = {
    "GPTDataset": GPTDataset,
    "BlendedGPTDataset": BlendedGPTDataset,
    "GPTSyntheticDataset": GPTSyntheticDataset,
    "SyntheticImageNetDataset": SyntheticImageNetDataset,
    "ImageFolderDataset": ImageFolderDataset,
    "GeneralClsDataset": ImageFolderDataset,
    "ContrativeLearningDataset": ContrativeLearningDataset,
    "CIFAR10": CIFAR10Dataset,
    "CIFAR10Dataset": CIFAR10Dataset,
    "ErnieSyntheticDataset": ErnieSyntheticDataset,
    "ErnieWWMDataset": ErnieWWMDataset,
}


def _register_lazy():
    from paddlefleetx_amd.data.eval_dataset import (Lambada_Eval_Dataset,
                                                    LM_Eval_Dataset)
    from paddlefleetx_amd.data.glue_dataset import (GLUEDataset,
                                                    SyntheticGLUEDataset)
    _DATASETS.setdefault("LM_Eval_Dataset", LM_Eval_Dataset)
    _DATASETS.setdefault("Lambada_Eval_Dataset", Lambada_Eval_Dataset)
    _DATASETS.setdefault("GLUEDataset", GLUEDataset)
    _DATASETS.setdefault("SyntheticGLUEDataset", SyntheticGLUEDataset)
    from paddlefleetx_amd.data.multimodal_dataset import \
        SyntheticImagenDataset
    _DATASETS.setdefault("SyntheticImagenDataset", SyntheticImagenDataset)
    from paddlefleetx_amd.data.multimodal_dataset import ImagenFileDataset
    _DATASETS.setdefault("ImagenDataset", ImagenFileDataset)
    _DATASETS.setdefault("ImagenFileDataset", ImagenFileDataset)
    from paddlefleetx_amd.data.folding_dataset import SyntheticFoldingDataset
    _DATASETS.setdefault("SyntheticFoldingDataset", SyntheticFoldingDataset)


_register_lazy()


def register_dataset(name, cls):
    _DATASETS[name] = cls


def gpt_collate_fn(samples):
    tokens = torch.stack([s[0] for s in samples])
    position_ids = torch.stack([s[1] for s in samples])
    labels = torch.stack([s[2] for s in samples])
    loss_mask = torch.stack([s[3] for s in samples])
    return tokens, position_ids, labels, loss_mask


def _worker_init(worker_id: int):
    seed = torch.initial_seed() % (2 ** 31)
    np.random.seed(seed + worker_id)


def build_dataloader(data_cfg, mode: str = "Train", consumed_samples: int = 0,
                     batch_size: Optional[int] = None) -> Optional[DataLoader]:
    """data_cfg may be the full config (with Data/Global sections) or just Data."""
    if "Data" in data_cfg:
        if batch_size is None:
            batch_size = data_cfg.get("Global", {}).get("local_batch_size")
        data_cfg = data_cfg["Data"]
    section = data_cfg.get(mode) if mode in data_cfg else data_cfg
    if section is None:
        return None
    ds_cfg = dict(section.get("dataset", {}))
    name = ds_cfg.pop("name", "GPTSyntheticDataset")
    if name not in _DATASETS:
        raise ValueError(f"unknown dataset {name}")
    ds_cfg.setdefault("mode", mode)
    try:
        dataset = _DATASETS[name](**ds_cfg)
    except TypeError:
        ds_cfg.pop("mode", None)
        dataset = _DATASETS[name](**ds_cfg)

    loader_cfg = dict(section.get("loader", {}))
    sampler_cfg = dict(section.get("sampler", {}))
    if batch_size is None:
        batch_size = sampler_cfg.get("batch_size", loader_cfg.get("batch_size", 1))
    batch_size = int(batch_size)
    sampler = GPTBatchSampler(
        dataset, batch_size,
        shuffle=bool(sampler_cfg.get("shuffle", False)),
        drop_last=bool(sampler_cfg.get("drop_last", True)),
        rank=get_data_world_rank(), num_replicas=get_data_world_size(),
        consumed_samples=consumed_samples)
    # dataset may carry its own collate (None -> torch default collate);
    # GPT datasets use the tuple-of-stacks gpt_collate_fn
    if hasattr(dataset, "collate_fn"):
        collate = dataset.collate_fn
    else:
        collate = gpt_collate_fn
    loader = DataLoader(
        dataset, batch_sampler=sampler,
        num_workers=int(loader_cfg.get("num_workers", 0)),
        pin_memory=bool(loader_cfg.get("use_shared_memory", False)),
        collate_fn=collate, worker_init_fn=_worker_init,
        persistent_workers=int(loader_cfg.get("num_workers", 0)) > 0)
    logger.info(f"dataloader[{mode}]: dataset={name} len={len(dataset)} "
                f"batch_size={batch_size} replicas={get_data_world_size()}")
    return loader
