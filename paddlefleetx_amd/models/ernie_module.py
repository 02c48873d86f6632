"""ERNIE modules: pretrain (MLM+NSP) and sequence classification.

Reference: ppfleetx/models/language_model/ernie/ernie_module.py
  ErnieModule :120 (batch = [input_ids, token_type_ids, position_ids,
  attention_mask(optional), masked_lm_labels, next_sentence_labels]),
  ErnieSeqClsModule :237.
"""

from __future__ import annotations

import torch

from paddlefleetx_amd.models.ernie import (ErnieForPretraining,
                                           ErnieForSequenceClassification,
                                           ErnieModel,
                                           ErniePretrainingCriterion)
from paddlefleetx_amd.models.language_module import (LanguageModule,
                                                     _model_dtype)
from paddlefleetx_amd.parallel.env import get_hcg


def _ernie_cfg(cfg) -> dict:
    mcfg = dict(cfg["Model"])
    for k in ("name", "module", "num_classes"):
        mcfg.pop(k, None)
    mp = get_hcg().get_model_parallel_world_size()
    if mp > 1:
        from paddlefleetx_amd.models.language_module import \
            vocab_size_with_padding
        mcfg["vocab_size"] = vocab_size_with_padding(
            mcfg.get("vocab_size", 18000),
            mcfg.pop("vocab_size_divisible_unit", 128), mp)
    return mcfg


def _ernie_model(cfg) -> ErnieModel:
    mcfg = _ernie_cfg(cfg)
    moe_configs = mcfg.pop("moe_configs", None)
    if moe_configs:
        hcg = get_hcg()
        assert hcg.get_pipe_parallel_world_size() == 1 and \
            hcg.get_sharding_parallel_world_size() == 1, \
            "ERNIE MoE requires pp==1 and sharding==1"
    return ErnieModel(dtype=_model_dtype(cfg), moe_configs=moe_configs, **mcfg)


class ErnieModule(LanguageModule):
    def get_model(self):
        hcg = get_hcg()
        if hcg.get_pipe_parallel_world_size() > 1:
            from paddlefleetx_amd.models.ernie.pipeline_model import \
                ErnieForPretrainingPipe
            cfg = self.configs
            vpp = int(cfg.get("Distributed", {}).get("pipeline", {})
                      .get("virtual_pp_degree", 1) or 1)
            mcfg = _ernie_cfg(cfg)
            mcfg.pop("moe_configs", None)
            return ErnieForPretrainingPipe(dtype=_model_dtype(cfg),
                                           virtual_pp_degree=vpp, **mcfg)
        return ErnieForPretraining(_ernie_model(self.configs))

    def get_loss_fn(self):
        if get_hcg().get_pipe_parallel_world_size() > 1:
            from paddlefleetx_amd.models.ernie.pipeline_model import \
                ErniePipeCriterion
            return ErniePipeCriterion()
        return ErniePretrainingCriterion(with_nsp_loss=True)

    def training_step(self, batch):
        input_ids, token_type_ids, masked_lm_labels, next_sentence_labels = \
            batch[:4]
        pred, seq_rel = self.model(input_ids, token_type_ids)
        mlm, nsp = self.loss_fn(pred, seq_rel, masked_lm_labels,
                                next_sentence_labels)
        loss = mlm + nsp
        # MoE gate aux loss, if any
        from paddlefleetx_amd.models.moe.moe_layer import MoELayer
        for m in self.model.modules():
            if isinstance(m, MoELayer):
                gl = m.gate.get_loss()
                if gl is not None:
                    loss = loss + 0.01 * gl.to(loss.dtype)
        return loss

    def validation_step(self, batch):
        return self.training_step(batch)


class ErnieSeqClsModule(LanguageModule):
    def get_model(self):
        num_classes = int(self.configs["Model"].get("num_classes", 2))
        return ErnieForSequenceClassification(_ernie_model(self.configs),
                                              num_classes=num_classes)

    def get_loss_fn(self):
        return torch.nn.CrossEntropyLoss()

    def training_step(self, batch):
        input_ids, token_type_ids, labels = batch[:3]
        logits = self.model(input_ids, token_type_ids)
        return self.loss_fn(logits.float(), labels)

    def validation_step(self, batch):
        input_ids, token_type_ids, labels = batch[:3]
        logits = self.model(input_ids, token_type_ids)
        self._last_acc = float((logits.argmax(-1) == labels).float().mean())
        return self.loss_fn(logits.float(), labels)
