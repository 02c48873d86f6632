"""GPT modules: pretrain / generation / eval.

Reference: ppfleetx/models/language_model/language_module.py
  LanguageModule (:73-146, ips log grammar :108-113), GPTModule (:148,
  model selection :181-192), GPTGenerationModule (:490), GPTEvalModule (:600).
"""

from __future__ import annotations

import time
from typing import Any, Dict

import torch

from paddlefleetx_amd.core.module import BasicModule
from paddlefleetx_amd.models.gpt.model import (GPTForPretraining, GPTModel,
                                               GPTPretrainingCriterion)
from paddlefleetx_amd.parallel.env import get_data_world_size, get_hcg
from paddlefleetx_amd.utils.log import logger


def vocab_size_with_padding(vocab_size: int, div_unit: int, mp_degree: int) -> int:
    """Pad vocab to a multiple of div_unit*mp (reference language_module.py:62)."""
    mult = div_unit * mp_degree
    return ((vocab_size + mult - 1) // mult) * mult


def _model_dtype(cfg) -> torch.dtype:
    mp = cfg.get("Engine", {}).get("mix_precision", {})
    if not mp.get("enable", True):
        return torch.float32
    return {"bfloat16": torch.bfloat16, "float16": torch.float16,
            "float32": torch.float32}[mp.get("dtype", "bfloat16")]


class LanguageModule(BasicModule):
    """Adds the train/val step hooks + the `ips:` throughput log line."""

    def __init__(self, configs):
        self.nranks = get_hcg().world_size
        super().__init__(configs)

    def training_step(self, batch):
        tokens, position_ids, labels, loss_mask = batch
        logits = self(tokens, position_ids)
        return self.loss_fn(logits, labels, loss_mask)

    def validation_step(self, batch):
        tokens, position_ids, labels, loss_mask = batch
        logits = self(tokens, position_ids)
        return self.loss_fn(logits, labels, loss_mask)

    def training_step_end(self, log_dict):
        speed = 1.0 / max(log_dict["train_cost"], 1e-12)
        cfg = self.configs
        gbs = cfg["Global"]["global_batch_size"]
        seq = cfg["Model"].get("max_position_embeddings", None) or \
            cfg["Data"].get("seq_len", 1024) if "Data" in cfg else 1024
        ips_total = speed * gbs * seq
        ips = ips_total / max(1, get_data_world_size())
        logger.train(
            "[train] epoch: %d, batch: %d, loss: %.9f, avg_batch_cost: %.5f sec, "
            "speed: %.2f step/s, ips_total: %.0f tokens/s, ips: %.0f tokens/s, "
            "learning rate: %.5e, found_inf: %.0f"
            % (log_dict["epoch"], log_dict["batch"], log_dict["loss"],
               log_dict["train_cost"], speed, ips_total, ips, log_dict["lr"],
               log_dict.get("found_inf", 0)))

    def validation_step_end(self, log_dict):
        speed = 1.0 / max(log_dict["eval_cost"], 1e-12)
        logger.eval("[eval] epoch: %d, batch: %d, loss: %.9f, "
                    "avg_eval_cost: %.5f sec, speed: %.2f step/s"
                    % (log_dict["epoch"], log_dict["batch"], log_dict["loss"],
                       log_dict["eval_cost"], speed))


class GPTModule(LanguageModule):
    """Pretraining module; picks single/hybrid/pipe network by topology
    (reference language_module.py:181-192)."""

    def get_model(self):
        cfg = self.configs
        mcfg = dict(cfg["Model"])
        for k in ("name", "moe_configs", "vocab_size_divisible_unit"):
            mcfg.pop(k, None)
        hcg = get_hcg()
        mp_deg = hcg.get_model_parallel_world_size()
        mcfg["vocab_size"] = vocab_size_with_padding(
            mcfg.get("vocab_size", 50304),
            cfg["Model"].get("vocab_size_divisible_unit", 128), mp_deg)
        dtype = _model_dtype(cfg)
        use_rec = mcfg.pop("use_recompute", False)
        mcfg["use_recompute"] = use_rec
        if hcg.get_pipe_parallel_world_size() > 1:
            from paddlefleetx_amd.models.gpt.pipeline_model import \
                GPTForPretrainingPipe
            pipe_cfg = cfg.get("Distributed", {}).get("pipeline", {})
            vpp = int(pipe_cfg.get("virtual_pp_degree", 1) or 1)
            psr = bool(pipe_cfg.get("enable_partial_send_recv", False))
            return GPTForPretrainingPipe(dtype=dtype, virtual_pp_degree=vpp,
                                         partial_send_recv=psr, **mcfg)
        return GPTForPretraining(GPTModel(dtype=dtype, **mcfg))

    def get_loss_fn(self):
        return GPTPretrainingCriterion()

    def pretreating_batch(self, batch):
        hcg = get_hcg()
        cp = hcg.get_context_parallel_world_size()
        if cp > 1:
            r = hcg.get_context_parallel_rank()
            mcfg = self.configs.get("Model", {})
            if mcfg.get("cp_backend") == "ring" and \
                    mcfg.get("cp_zigzag", False):
                # zigzag sharding: half-chunks (r, 2cp-1-r) level the
                # causal work across the ring (parallel/ring.py)
                from paddlefleetx_amd.parallel.ring import zigzag_slice
                batch = tuple(
                    zigzag_slice(t, cp, r, dim=1)
                    if torch.is_tensor(t) and t.ndim >= 2 else t
                    for t in batch)
            else:
                # each cp rank takes its sequence chunk (Ulysses/ring
                # sequential sharding)
                batch = tuple(
                    torch.chunk(t, cp, dim=1)[r].contiguous()
                    if torch.is_tensor(t) and t.ndim >= 2 else t
                    for t in batch)
        return batch


class GPTFinetuneModule(LanguageModule):
    """GLUE finetuning (reference language_module.py:228-488): GPT backbone
    + sequence-classification head, per-task metric, CE or MSE loss."""

    def __init__(self, configs):
        from paddlefleetx_amd.data.glue_dataset import GLUE_METRICS
        self.task = configs["Model"].get("task", "sst2").lower()
        metric_name = configs["Model"].get("metric", None) or \
            GLUE_METRICS.get(self.task, "Accuracy")
        from paddlefleetx_amd.models import metrics as M
        self.metric = getattr(M, metric_name)()
        self.regression = self.task == "stsb"
        super().__init__(configs)

    def get_model(self):
        cfg = self.configs
        mcfg = dict(cfg["Model"])
        for k in ("name", "task", "metric", "num_classes",
                  "vocab_size_divisible_unit", "moe_configs"):
            mcfg.pop(k, None)
        hcg = get_hcg()
        mcfg["vocab_size"] = vocab_size_with_padding(
            mcfg.get("vocab_size", 50304),
            cfg["Model"].get("vocab_size_divisible_unit", 128),
            hcg.get_model_parallel_world_size())
        num_classes = int(cfg["Model"].get("num_classes",
                                           1 if self.regression else 2))
        from paddlefleetx_amd.models.gpt.model import (
            GPTForSequenceClassification, GPTModel)
        gpt = GPTModel(dtype=_model_dtype(cfg), **mcfg)
        return GPTForSequenceClassification(gpt, num_classes=num_classes)

    def get_loss_fn(self):
        if self.regression:
            return torch.nn.MSELoss()
        return torch.nn.CrossEntropyLoss()

    def training_step(self, batch):
        ids, mask, labels = batch[:3]
        logits = self.model(ids, attention_mask=mask)
        if self.regression:
            return self.loss_fn(logits.float().squeeze(-1), labels.float())
        return self.loss_fn(logits.float(), labels)

    def validation_step(self, batch):
        ids, mask, labels = batch[:3]
        logits = self.model(ids, attention_mask=mask)
        if self.regression:
            self.metric.update(logits.squeeze(-1), labels)
            return self.loss_fn(logits.float().squeeze(-1), labels.float())
        self.metric.update(logits, labels)
        return self.loss_fn(logits.float(), labels)

    def validation_step_end(self, log_dict):
        vals = self.metric.accumulate()
        names = self.metric.name()
        if not isinstance(vals, tuple):
            vals, names = (vals,), (names,)
        stats = ", ".join(f"{n}: {v:.4f}" for n, v in zip(names, vals))
        logger.eval("[eval] epoch: %d, batch: %d, loss: %.9f, %s"
                    % (log_dict["epoch"], log_dict["batch"],
                       log_dict["loss"], stats))


class GPTGenerationModule(BasicModule):
    """Text generation (reference language_module.py:490-598)."""

    def __init__(self, configs):
        super().__init__(configs)
        gcfg = configs.get("Generation", {})
        self.top_k = gcfg.get("top_k", 0)
        self.top_p = gcfg.get("top_p", 1.0)
        self.temperature = gcfg.get("temperature", 1.0)
        self.max_dec_len = gcfg.get("max_dec_len", 64)
        self.use_topp_sampling = gcfg.get("use_topp_sampling", False)

    def get_model(self):
        cfg = self.configs
        mcfg = dict(cfg["Model"])
        for k in ("name", "moe_configs", "vocab_size_divisible_unit"):
            mcfg.pop(k, None)
        hcg = get_hcg()
        mcfg["vocab_size"] = vocab_size_with_padding(
            mcfg.get("vocab_size", 50304),
            cfg["Model"].get("vocab_size_divisible_unit", 128),
            hcg.get_model_parallel_world_size())
        from paddlefleetx_amd.models.gpt.generation import GPTForGeneration
        dtype = _model_dtype(cfg)
        gpt = GPTModel(dtype=dtype, **{k: v for k, v in mcfg.items()
                                       if k != "use_recompute"})
        return GPTForGeneration(gpt, self.configs.get("Generation", {}))

    def generate(self, input_ids: torch.Tensor):
        return self.model(input_ids)
