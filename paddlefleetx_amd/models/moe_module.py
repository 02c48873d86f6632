"""MoE pretraining module.

Reference: ppfleetx/models/language_model/language_module.py:736 MoEModule —
GPT network with expert_mode decoder layers; training loss = CE +
balance_loss_weight * sum of per-layer gate aux losses.
"""

from __future__ import annotations

import torch

from paddlefleetx_amd.models.gpt.model import (GPTForPretraining, GPTModel,
                                               GPTPretrainingCriterion)
from paddlefleetx_amd.models.language_module import (LanguageModule,
                                                     _model_dtype,
                                                     vocab_size_with_padding)
from paddlefleetx_amd.models.moe.moe_layer import MoELayer
from paddlefleetx_amd.parallel.env import get_hcg


class MoEModule(LanguageModule):
    def __init__(self, configs):
        mcfg = configs["Model"].get("moe_configs", {}) or {}
        self.balance_loss_weight = float(mcfg.get("balance_loss_weight", 0.01))
        super().__init__(configs)

    def get_model(self):
        cfg = self.configs
        mcfg = dict(cfg["Model"])
        moe_configs = dict(mcfg.pop("moe_configs", {}) or {})
        moe_configs.setdefault("expert_mode", True)
        for k in ("name", "vocab_size_divisible_unit"):
            mcfg.pop(k, None)
        hcg = get_hcg()
        assert hcg.get_pipe_parallel_world_size() == 1 and \
            hcg.get_sharding_parallel_world_size() == 1, \
            "MoE requires pp==1 and sharding==1 (comm_groups.py:133-137)"
        mcfg["vocab_size"] = vocab_size_with_padding(
            mcfg.get("vocab_size", 50304),
            cfg["Model"].get("vocab_size_divisible_unit", 128),
            hcg.get_model_parallel_world_size())
        return GPTForPretraining(GPTModel(
            dtype=_model_dtype(cfg), moe_configs=moe_configs, **mcfg))

    def get_loss_fn(self):
        return GPTPretrainingCriterion()

    def _gate_loss(self):
        losses = []
        for m in self.model.modules():
            if isinstance(m, MoELayer):
                gl = m.gate.get_loss()
                if gl is not None:
                    losses.append(gl)
        if not losses:
            return None
        return torch.stack(losses).sum()

    def training_step(self, batch):
        tokens, position_ids, labels, loss_mask = batch
        logits = self(tokens, position_ids)
        loss = self.loss_fn(logits, labels, loss_mask)
        gate_loss = self._gate_loss()
        if gate_loss is not None:
            loss = loss + self.balance_loss_weight * gate_loss.to(loss.dtype)
        return loss
