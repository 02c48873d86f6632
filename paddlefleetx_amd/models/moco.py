"""MoCo v1/v2 contrastive pretraining + linear classification.

Reference: ppfleetx/models/vision_model/moco/moco.py — concat_all_gather
:36, MoCoV2Projector :50, MoCoClassifier :70, MoCo :94 (momentum encoder
:136, queue :147, shuffle-BN via all_gather+broadcast+index_select
:162-206, InfoNCE logits :208-243). RCCL collectives via
torch.distributed.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.core.module import BasicModule
from paddlefleetx_amd.utils.log import logger


@torch.no_grad()
def concat_all_gather(tensor: torch.Tensor) -> torch.Tensor:
    if not dist.is_initialized() or dist.get_world_size() < 2:
        return tensor
    out = [torch.empty_like(tensor) for _ in range(dist.get_world_size())]
    dist.all_gather(out, tensor.contiguous())
    return torch.cat(out, dim=0)


class MoCoV2Projector(nn.Module):
    def __init__(self, with_pool: bool, in_dim: int, out_dim: int):
        super().__init__()
        self.with_pool = with_pool
        if with_pool:
            self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.mlp = nn.Sequential(nn.Linear(in_dim, out_dim), nn.ReLU())

    def forward(self, x):
        if self.with_pool:
            x = torch.flatten(self.avgpool(x), 1)
        return self.mlp(x)


class MoCoClassifier(nn.Module):
    def __init__(self, with_pool: bool, num_features: int, num_classes: int):
        super().__init__()
        self.with_pool = with_pool
        if with_pool:
            self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(num_features, num_classes)

    def forward(self, x):
        if self.with_pool:
            x = torch.flatten(self.avgpool(x), 1)
        return self.fc(x)


class MoCo(nn.Module):
    def __init__(self, base_encoder: nn.Module, base_projector: nn.Module,
                 base_classifier: nn.Module, momentum_encoder: nn.Module,
                 momentum_projector: nn.Module,
                 momentum_classifier: nn.Module, dim: int = 128,
                 K: int = 65536, m: float = 0.999, T: float = 0.07,
                 **unused):
        super().__init__()
        self.m, self.T, self.K = m, T, K
        self.base_encoder = nn.Sequential(base_encoder, base_projector,
                                          base_classifier)
        self.momentum_encoder = nn.Sequential(momentum_encoder,
                                              momentum_projector,
                                              momentum_classifier)
        for pb, pm in zip(self.base_encoder.parameters(),
                          self.momentum_encoder.parameters()):
            pm.data.copy_(pb.data)
            pm.requires_grad = False
        self.register_buffer("queue", F.normalize(torch.randn(dim, K), dim=0))
        self.register_buffer("queue_ptr", torch.zeros(1, dtype=torch.long))

    @torch.no_grad()
    def _update_momentum_encoder(self):
        for pb, pm in zip(self.base_encoder.parameters(),
                          self.momentum_encoder.parameters()):
            pm.data.mul_(self.m).add_(pb.data, alpha=1.0 - self.m)

    @torch.no_grad()
    def _dequeue_and_enqueue(self, keys):
        keys = concat_all_gather(keys)
        bs = keys.shape[0]
        ptr = int(self.queue_ptr[0])
        assert self.K % bs == 0, "queue size must divide global batch"
        self.queue[:, ptr:ptr + bs] = keys.T
        self.queue_ptr[0] = (ptr + bs) % self.K

    @torch.no_grad()
    def _batch_shuffle_ddp(self, x):
        bs_this = x.shape[0]
        x_gather = concat_all_gather(x)
        bs_all = x_gather.shape[0]
        num_gpus = bs_all // bs_this
        idx_shuffle = torch.randperm(bs_all, device=x.device)
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.broadcast(idx_shuffle, src=0)
        idx_unshuffle = torch.argsort(idx_shuffle)
        rank = dist.get_rank() if dist.is_initialized() else 0
        idx_this = idx_shuffle.view(num_gpus, -1)[rank]
        return x_gather[idx_this], idx_unshuffle

    @torch.no_grad()
    def _batch_unshuffle_ddp(self, x, idx_unshuffle):
        bs_this = x.shape[0]
        x_gather = concat_all_gather(x)
        num_gpus = x_gather.shape[0] // bs_this
        rank = dist.get_rank() if dist.is_initialized() else 0
        idx_this = idx_unshuffle.view(num_gpus, -1)[rank]
        return x_gather[idx_this]

    def forward(self, x1, x2):
        q = F.normalize(self.base_encoder(x1), dim=1)
        with torch.no_grad():
            self._update_momentum_encoder()
            k, idx_unshuffle = self._batch_shuffle_ddp(x2)
            k = F.normalize(self.momentum_encoder(k), dim=1)
            k = self._batch_unshuffle_ddp(k, idx_unshuffle)
        l_pos = (q * k).sum(dim=1, keepdim=True)
        l_neg = q @ self.queue.clone().detach()
        logits = torch.cat([l_pos, l_neg], dim=1) / self.T
        labels = torch.zeros(logits.shape[0], dtype=torch.long,
                             device=logits.device)
        self._dequeue_and_enqueue(k)
        return logits, labels


class MOCOModule(BasicModule):
    """MoCo v1/v2 pretraining module (general_moco_module in reference)."""

    def get_model(self):
        mcfg = dict(self.configs["Model"].get("model", {}))
        backbone = mcfg.get("backbone", "resnet50")
        dim = int(mcfg.get("dim", 128))
        K = int(mcfg.get("K", 65536))
        m = float(mcfg.get("m", 0.999))
        T = float(mcfg.get("T", 0.07))
        v2 = bool(mcfg.get("v2", True))
        from paddlefleetx_amd.models import resnet as R
        make = getattr(R, backbone)

        def enc():
            return make(class_num=0, with_pool=False)
        feats = enc().num_features
        if v2:
            projs = (MoCoV2Projector(True, feats, feats),
                     MoCoV2Projector(True, feats, feats))
            clfs = (MoCoClassifier(False, feats, dim),
                    MoCoClassifier(False, feats, dim))
        else:
            projs = (nn.Identity(), nn.Identity())
            clfs = (MoCoClassifier(True, feats, dim),
                    MoCoClassifier(True, feats, dim))
        return MoCo(enc(), projs[0], clfs[0], enc(), projs[1], clfs[1],
                    dim=dim, K=K, m=m, T=T)

    def get_loss_fn(self):
        return nn.CrossEntropyLoss()

    def training_step(self, batch):
        # batch: ((view1, view2), _) from a two-crop transform, or a single
        # image tensor (both views identical — plumbing tests)
        if isinstance(batch[0], (tuple, list)):
            x1, x2 = batch[0]
        else:
            x1 = x2 = batch[0]
        logits, labels = self.model(x1, x2)
        return self.loss_fn(logits.float(), labels)

    def validation_step(self, batch):
        return self.training_step(batch)

    def training_step_end(self, log_dict):
        logger.train("[train] epoch: %d, batch: %d, loss: %.9f, "
                     "avg_batch_cost: %.5f sec"
                     % (log_dict["epoch"], log_dict["batch"],
                        log_dict["loss"], log_dict["train_cost"]))
