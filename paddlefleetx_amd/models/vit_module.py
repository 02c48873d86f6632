"""Vision classification module (ViT / general).

Reference: ppfleetx/models/vision_model/general_classification_module.py:31
GeneralClsModule — builds the network from Model.model config, CE (or
soft-target) loss, TopK accuracy on eval, ips log in images/s.
"""

from __future__ import annotations

import time
from typing import Any, Dict

import torch
import torch.nn.functional as F

from paddlefleetx_amd.core.module import BasicModule
from paddlefleetx_amd.parallel.env import get_data_world_size
from paddlefleetx_amd.utils.log import logger


def topk_accuracy(logits: torch.Tensor, labels: torch.Tensor,
                  ks=(1, 5)) -> Dict[str, float]:
    maxk = max(ks)
    _, pred = logits.topk(maxk, dim=-1)
    correct = pred.eq(labels.unsqueeze(-1))
    return {f"top{k}": float(correct[:, :k].any(dim=-1).float().mean())
            for k in ks}


class SoftTargetCrossEntropy(torch.nn.Module):
    """CE against soft targets (mixup/cutmix labels)."""

    def forward(self, logits, target):
        return (-target * F.log_softmax(logits.float(), dim=-1)).sum(-1).mean()


class GeneralClsModule(BasicModule):
    def __init__(self, configs):
        super().__init__(configs)
        self.acc_ks = tuple(configs["Model"].get("metric", {})
                            .get("topk", (1, 5)))

    def get_model(self):
        mcfg = dict(self.configs["Model"])
        net_cfg = dict(mcfg.get("model", {}))
        name = net_cfg.pop("name", "ViT_base_patch16_224")
        mp = self.configs.get("Engine", {}).get("mix_precision", {})
        dtype = {"bfloat16": torch.bfloat16, "float16": torch.float16,
                 "float32": torch.float32}[mp.get("dtype", "bfloat16")] \
            if mp.get("enable", True) else torch.float32
        from paddlefleetx_amd.models.vit import build_vit
        return build_vit(name, dtype=dtype, **net_cfg)

    def get_loss_fn(self):
        lcfg = self.configs["Model"].get("loss", {})
        name = lcfg.get("train", {}).get("name", "CELoss") \
            if isinstance(lcfg.get("train"), dict) else "CELoss"
        if name == "SoftTargetCELoss":
            return SoftTargetCrossEntropy()
        return torch.nn.CrossEntropyLoss()

    def training_step(self, batch):
        images, labels = batch
        logits = self(images)
        if isinstance(self.loss_fn, torch.nn.CrossEntropyLoss):
            return self.loss_fn(logits.float(), labels)
        return self.loss_fn(logits, labels)

    def validation_step(self, batch):
        images, labels = batch
        logits = self(images)
        self._last_acc = topk_accuracy(logits.float(), labels, self.acc_ks)
        return F.cross_entropy(logits.float(), labels)

    def training_step_end(self, log_dict):
        speed = 1.0 / max(log_dict["train_cost"], 1e-12)
        gbs = self.configs["Global"]["global_batch_size"]
        ips_total = speed * gbs
        ips = ips_total / max(1, get_data_world_size())
        logger.train(
            "[train] epoch: %d, batch: %d, loss: %.9f, avg_batch_cost: %.5f "
            "sec, speed: %.2f step/s, ips_total: %.0f images/s, ips: %.0f "
            "images/s, learning rate: %.5e"
            % (log_dict["epoch"], log_dict["batch"], log_dict["loss"],
               log_dict["train_cost"], speed, ips_total, ips, log_dict["lr"]))

    def validation_step_end(self, log_dict):
        acc = getattr(self, "_last_acc", {})
        accs = ", ".join(f"{k}: {v:.4f}" for k, v in acc.items())
        logger.eval("[eval] epoch: %d, batch: %d, loss: %.9f, %s"
                    % (log_dict["epoch"], log_dict["batch"],
                       log_dict["loss"], accs))
