"""Offline eval module: WikiText PPL / LAMBADA cloze accuracy.

Reference: ppfleetx/models/language_model/language_module.py GPTEvalModule
:600-734 — Offline_Eval config section selects LM_Eval_Dataset
(cloze_eval=False: summed masked CE -> ppl and adjusted ppl) or
Lambada_Eval_Dataset (cloze_eval=True: whole-word exact-match count ->
accuracy).
"""

from __future__ import annotations

import math

import torch

from paddlefleetx_amd.core.module import BasicModule
from paddlefleetx_amd.models.language_module import (_model_dtype,
                                                     vocab_size_with_padding)
from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.utils.log import logger


class GPTEvalModule(BasicModule):
    def __init__(self, configs):
        self.eval_cfgs = configs.get("Offline_Eval", {})
        self.cloze_eval = bool(self.eval_cfgs.get("cloze_eval", False))
        self.total_score = 0.0
        self.first_step = True
        self.num_original_tokens = None
        self.num_tokenized_tokens = None
        self.num_examples = None
        super().__init__(configs)

    def get_model(self):
        cfg = self.configs
        mcfg = dict(cfg["Model"])
        for k in ("name", "moe_configs", "vocab_size_divisible_unit"):
            mcfg.pop(k, None)
        mcfg["vocab_size"] = vocab_size_with_padding(
            mcfg.get("vocab_size", 50304),
            cfg["Model"].get("vocab_size_divisible_unit", 128),
            get_hcg().get_model_parallel_world_size())
        from paddlefleetx_amd.models.gpt.model import (GPTForPretraining,
                                                       GPTModel)
        return GPTForPretraining(GPTModel(dtype=_model_dtype(cfg), **mcfg))

    def validation_step(self, batch):
        tokens, position_ids, labels, loss_mask, info = batch
        logits = self.model(tokens, position_ids)
        if self.first_step:
            if not self.cloze_eval:
                self.num_original_tokens = int(info[0][0])
                self.num_tokenized_tokens = int(info[0][1])
            else:
                self.num_examples = int(info[0][0])
            self.first_step = False
        if not self.cloze_eval:
            # summed masked CE (language_module.py:672-676)
            ce = torch.nn.functional.cross_entropy(
                logits.float().reshape(-1, logits.shape[-1]),
                labels.reshape(-1), reduction="none")
            score = (ce * loss_mask.reshape(-1).float()).sum()
        else:
            preds = logits.argmax(-1)
            correct = (preds == labels) | ~loss_mask.bool()
            score = correct.all(dim=-1).float().sum()
        self.total_score += float(score)
        return score

    def validation_step_end(self, log_dict):
        name = "number correct" if self.cloze_eval else "loss"
        logger.eval("[eval] epoch: %d, batch: %d, %s: %.9f"
                    % (log_dict["epoch"], log_dict["batch"], name,
                       self.total_score))

    def validation_epoch_end(self):
        """Final summary (language_module.py:706-729)."""
        if not self.cloze_eval:
            total_loss = self.total_score / max(1,
                                                self.num_tokenized_tokens - 1)
            ppl = math.exp(min(20, total_loss))
            token_ratio = (self.num_tokenized_tokens - 1) / max(
                1, self.num_original_tokens - 1)
            adjusted = math.exp(min(20, total_loss * token_ratio))
            logger.info(f"validation results | avg loss: {total_loss:.4E} | "
                        f"ppl: {ppl:.4E} | adjusted ppl: {adjusted:.4E} | "
                        f"token ratio: {token_ratio}")
            return {"loss": total_loss, "ppl": ppl, "adjusted_ppl": adjusted}
        acc = self.total_score / max(1, self.num_examples)
        logger.info(f"validation results | number correct: "
                    f"{self.total_score:.4E} | total examples: "
                    f"{self.num_examples} | avg accuracy: {acc:.4E}")
        return {"number_correct": self.total_score, "acc": acc}
