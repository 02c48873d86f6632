"""Logits processors for generation.

Reference: ppfleetx/models/language_model/gpt/dygraph/processor.py
  LogitsProcessorList :22, MinLengthLogitsProcessor :48,
  RepetitionPenaltyLogitsProcessor :77, HammingDiversityLogitsProcessor :106,
  ForcedBOSTokenLogitsProcessor :158, ForcedEOSTokenLogitsProcessor :180.
"""

from __future__ import annotations

from typing import List

import torch


class LogitsProcessor:
    def __call__(self, input_ids: torch.Tensor, logits: torch.Tensor
                 ) -> torch.Tensor:
        raise NotImplementedError


class LogitsProcessorList(list):
    def __call__(self, input_ids: torch.Tensor, logits: torch.Tensor,
                 **kwargs) -> torch.Tensor:
        for proc in self:
            logits = proc(input_ids, logits)
        return logits


class MinLengthLogitsProcessor(LogitsProcessor):
    """Force eos probability to -inf until min_length is reached."""

    def __init__(self, min_length: int, eos_token_id: int):
        if not isinstance(min_length, int) or min_length < 0:
            raise ValueError(f"min_length must be a non-negative int, got {min_length}")
        if not isinstance(eos_token_id, int) or eos_token_id < 0:
            raise ValueError(f"eos_token_id must be a non-negative int, got {eos_token_id}")
        self.min_length = min_length
        self.eos_token_id = eos_token_id

    def __call__(self, input_ids, logits):
        if input_ids.shape[-1] < self.min_length:
            logits = logits.clone()
            logits[:, self.eos_token_id] = -float("inf")
        return logits


class RepetitionPenaltyLogitsProcessor(LogitsProcessor):
    """CTRL-style repetition penalty (https://arxiv.org/abs/1909.05858)."""

    def __init__(self, penalty: float):
        if not (isinstance(penalty, (int, float)) and penalty > 0):
            raise ValueError(f"penalty must be > 0, got {penalty}")
        self.penalty = float(penalty)

    def __call__(self, input_ids, logits):
        score = torch.gather(logits, 1, input_ids)
        score = torch.where(score < 0, score * self.penalty,
                            score / self.penalty)
        logits = logits.clone()
        logits.scatter_(1, input_ids, score)
        return logits


class HammingDiversityLogitsProcessor(LogitsProcessor):
    """Diverse beam search penalty (only active during group beam search)."""

    def __init__(self, diversity_rate: float, num_beams: int,
                 num_beam_groups: int):
        self._diversity_rate = float(diversity_rate)
        self._num_beams = num_beams
        self._num_sub_beams = num_beams // num_beam_groups

    def __call__(self, input_ids, logits, current_tokens=None,
                 beam_group_idx: int = 0):
        if current_tokens is None:
            return logits
        batch_size = current_tokens.shape[0] // self._num_beams
        group_start = beam_group_idx * self._num_sub_beams
        group_end = min(group_start + self._num_sub_beams, self._num_beams)
        group_size = group_end - group_start
        vocab_size = logits.shape[-1]
        if group_start == 0:
            return logits
        logits = logits.clone()
        for batch_idx in range(batch_size):
            prev = current_tokens[batch_idx * self._num_beams:
                                  batch_idx * self._num_beams + group_start]
            freq = torch.bincount(prev, minlength=vocab_size).to(logits.dtype)
            logits[batch_idx * group_size:(batch_idx + 1) * group_size] -= \
                self._diversity_rate * freq
        return logits


class ForcedBOSTokenLogitsProcessor(LogitsProcessor):
    def __init__(self, bos_token_id: int):
        self.bos_token_id = bos_token_id

    def __call__(self, input_ids, logits):
        if input_ids.shape[-1] == 1:
            logits = torch.full_like(logits, -float("inf"))
            logits[:, self.bos_token_id] = 0.0
        return logits


class ForcedEOSTokenLogitsProcessor(LogitsProcessor):
    def __init__(self, max_length: int, eos_token_id: int):
        self.max_length = max_length
        self.eos_token_id = eos_token_id

    def __call__(self, input_ids, logits):
        if input_ids.shape[-1] == self.max_length - 1:
            logits = torch.full_like(logits, -float("inf"))
            logits[:, self.eos_token_id] = 0.0
        return logits


def get_logits_processor(min_length: int = None, eos_token_id: int = None,
                         repetition_penalty: float = None,
                         forced_bos_token_id: int = None,
                         forced_eos_token_id: int = None,
                         max_length: int = None) -> LogitsProcessorList:
    """Assemble the processor list (reference single_model.py:1396-1405)."""
    processors = LogitsProcessorList()
    if min_length is not None and eos_token_id is not None and min_length > 0:
        processors.append(MinLengthLogitsProcessor(min_length, eos_token_id))
    if repetition_penalty is not None and repetition_penalty != 1.0:
        processors.append(RepetitionPenaltyLogitsProcessor(repetition_penalty))
    if forced_bos_token_id is not None:
        processors.append(ForcedBOSTokenLogitsProcessor(forced_bos_token_id))
    if forced_eos_token_id is not None and max_length is not None:
        processors.append(ForcedEOSTokenLogitsProcessor(max_length,
                                                        forced_eos_token_id))
    return processors
