"""GPT text generation: KV-cache decode loop + top-k/top-p sampling.

Reference: ppfleetx/models/language_model/gpt/dygraph/single_model.py
  GPTForGeneration :898 (sample :1139, TopKProcess :1150, TopPProcess :1158,
  custom top-p op hook :1240-1252, greedy/sampling dispatch forward :1322-1418)
and the hybrid twin hybrid_model.py:1209.

MI355X-native notes: the decode loop runs the same GPTModel with per-layer
KV caches ([B, H, S, D] tensors grown by concat); sampling post-processing
uses the hand-written gfx950 top-p kernel (csrc/topp.hip) when
`use_topp_sampling` is on, replacing the reference's CUB-based
ppfleetx/ops/topp_sampling.cu.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.models.gpt.model import GPTModel
from paddlefleetx_amd.models.gpt.processor import get_logits_processor
from paddlefleetx_amd.ops import topp_sampling
from paddlefleetx_amd.parallel.tp import parallel_matmul


def TopKProcess(probs: torch.Tensor, top_k: int, min_tokens_to_keep: int = 1
                ) -> torch.Tensor:
    """Zero all probability mass outside the top-k (single_model.py:1150-1156)."""
    top_k = min(max(top_k, min_tokens_to_keep), probs.shape[-1])
    topk_probs, _ = torch.topk(probs, k=top_k, dim=-1)
    kth = topk_probs[:, -1:].expand_as(probs)
    return torch.where(probs >= kth, probs, torch.zeros_like(probs))


def TopPProcess(probs: torch.Tensor, top_p: float, min_tokens_to_keep: int = 1
                ) -> torch.Tensor:
    """Nucleus filtering on the host graph (single_model.py:1158-1185)."""
    sorted_probs, sorted_indices = torch.sort(probs, descending=True, dim=-1)
    cumulative = torch.cumsum(sorted_probs, dim=-1)
    # remove tokens with cumulative prob above top_p, keeping at least one
    sorted_to_remove = cumulative > top_p
    sorted_to_remove[:, 0] = False
    if min_tokens_to_keep > 1:
        sorted_to_remove[:, :min_tokens_to_keep] = False
    # shift right: the first token crossing the threshold stays
    sorted_to_remove[:, 1:] = sorted_to_remove[:, :-1].clone()
    sorted_to_remove[:, 0] = False
    to_remove = sorted_to_remove.scatter(1, sorted_indices, sorted_to_remove)
    return probs.masked_fill(to_remove, 0.0)


class GPTForGeneration(nn.Module):
    """Decode-loop wrapper over GPTModel (single_model.py:898-1418)."""

    def __init__(self, gpt: GPTModel, configs: Optional[dict] = None):
        super().__init__()
        self.gpt = gpt
        cfg = configs or {}
        self.max_length = int(cfg.get("max_dec_len", 20))
        self.min_length = int(cfg.get("min_dec_len", 0))
        self.decode_strategy = cfg.get("decoding_strategy", "sampling")
        self.temperature = float(cfg.get("temperature", 1.0))
        self.top_k = int(cfg.get("top_k", 0))
        self.top_p = float(cfg.get("top_p", 1.0))
        self.repetition_penalty = float(cfg.get("repetition_penalty", 1.0))
        self.use_topp_sampling = bool(cfg.get("use_topp_sampling", False))
        self.eos_token_id = int(cfg.get("eos_token_id", 50256))
        self.pad_token_id = int(cfg.get("pad_token_id", 0))
        self.num_return_sequences = int(cfg.get("num_return_sequences", 1))

    # -- one forward --------------------------------------------------------
    def _logits(self, input_ids, position_ids, caches):
        hidden, new_caches = self.gpt(input_ids, position_ids, caches=caches,
                                      use_cache=True)
        logits = parallel_matmul(
            hidden[:, -1, :], self.gpt.embeddings.word_embeddings.weight,
            parallel_output=False)
        return logits.float(), new_caches

    def sample(self, input_ids: torch.Tensor,
               logits_processors=None) -> torch.Tensor:
        """Autoregressive sampling loop (single_model.py:1139-1320)."""
        B, prompt_len = input_ids.shape
        device = input_ids.device
        processors = logits_processors if logits_processors is not None else \
            get_logits_processor(
                min_length=prompt_len + self.min_length,
                eos_token_id=self.eos_token_id,
                repetition_penalty=self.repetition_penalty)

        position_ids = torch.arange(prompt_len, device=device).unsqueeze(0) \
            .expand(B, -1)
        # prefill: full prompt builds the KV cache (single_model.py:1285)
        logits, caches = self._logits(input_ids, position_ids, None)

        unfinished = torch.ones(B, dtype=torch.bool, device=device)
        all_ids = input_ids
        cur_len = prompt_len
        while cur_len < prompt_len + self.max_length:
            logits = processors(all_ids, logits)
            if self.decode_strategy == "greedy_search":
                next_tokens = logits.argmax(dim=-1, keepdim=True)
            else:
                if self.temperature != 1.0:
                    logits = logits / self.temperature
                probs = F.softmax(logits, dim=-1)
                if self.top_k > 0:
                    probs = TopKProcess(probs, self.top_k)
                if self.use_topp_sampling and self.top_p < 1.0:
                    tp = torch.full((B,), self.top_p, device=device)
                    next_tokens, _ = topp_sampling(probs, tp)
                else:
                    if self.top_p < 1.0:
                        probs = TopPProcess(probs, self.top_p)
                    next_tokens = torch.multinomial(probs, num_samples=1)
            next_tokens = torch.where(
                unfinished.unsqueeze(1), next_tokens,
                torch.full_like(next_tokens, self.pad_token_id))
            all_ids = torch.cat([all_ids, next_tokens], dim=-1)
            unfinished = unfinished & (next_tokens.squeeze(1) != self.eos_token_id)
            cur_len += 1
            if not unfinished.any():
                break
            pos = torch.full((B, 1), cur_len - 1, device=device,
                             dtype=torch.long)
            logits, caches = self._logits(next_tokens, pos, caches)
        return all_ids[:, prompt_len:]

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        was_training = self.gpt.training
        self.gpt.eval()
        with torch.no_grad():
            if self.decode_strategy == "sampling" and \
                    self.num_return_sequences > 1:
                # expand_inputs_for_generation (single_model.py:1408-1412):
                # each prompt row is repeated n times, so the output is
                # [B*n, max_dec_len] grouped by prompt
                input_ids = input_ids.repeat_interleave(
                    self.num_return_sequences, dim=0)
            out = self.sample(input_ids)
        if was_training:
            self.gpt.train()
        return out
