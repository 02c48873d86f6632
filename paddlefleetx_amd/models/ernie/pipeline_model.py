"""ERNIE as a pipeline of LayerDescs.

Reference: ppfleetx/models/language_model/ernie/dygraph/hybrid_model.py
  ErnieForPretrainingPipe :796 (EmbeddingPipe/EncoderLayerPipe descs,
  tied MLM decoder via SharedLayerDesc, NSP pooler on the last stage).

Stage boundary activation is the single hidden tensor [micro_b, S, H];
the last stage computes both heads locally (MLM scores + NSP logits) and
the pipe criterion reduces them to one scalar. Pretraining pipe runs
without a padding mask (fixed-length MLM batches), matching the
reference's pretrain data path. Composes with mp>1: the encoder layers
are the TP-aware ones from model.py.
"""

from __future__ import annotations

from typing import Any, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.models.ernie.model import (ErnieEmbeddings,
                                                 ErnieEncoderLayer,
                                                 ErniePooler,
                                                 ErniePretrainingCriterion,
                                                 ErniePretrainingHeads)
from paddlefleetx_amd.parallel.pp import (LayerDesc, PipelineModule,
                                          SharedLayerDesc)


class ErnieEmbeddingPipe(ErnieEmbeddings):
    """forward(input_ids, token_type_ids) -> hidden."""

    def forward(self, input_ids, token_type_ids=None):
        return super().forward(input_ids, token_type_ids=token_type_ids)


class ErnieEncoderLayerPipe(ErnieEncoderLayer):
    def forward(self, x):
        return super().forward(x, None)


class ErnieHeadsPipe(nn.Module):
    """Last stage: pooler + MLM/NSP heads. The MLM decoder reads
    `word_embeddings.weight` at FORWARD time so the SharedLayerDesc
    machinery (which swaps that attribute for stage 0's tensor) keeps the
    tie intact — same pattern as GPT's TiedLogitsPipe."""

    def __init__(self, vocab_size: int, hidden_size: int,
                 hidden_act: str = "gelu",
                 dtype: Optional[torch.dtype] = None, **unused: Any):
        super().__init__()
        from paddlefleetx_amd.models.ernie.model import _mp_degree
        self.mp = _mp_degree()
        if self.mp > 1:
            from paddlefleetx_amd.parallel.tp import VocabParallelEmbedding
            self.word_embeddings = VocabParallelEmbedding(
                vocab_size, hidden_size, dtype=dtype)
        else:
            self.word_embeddings = nn.Embedding(vocab_size, hidden_size,
                                                dtype=dtype)
        self.pooler = ErniePooler(hidden_size, dtype=dtype)
        self.transform = nn.Linear(hidden_size, hidden_size, dtype=dtype)
        self.activation = getattr(F, hidden_act)
        from paddlefleetx_amd.ops import FusedLayerNorm
        self.layer_norm = FusedLayerNorm(hidden_size, eps=1e-12, dtype=dtype)
        bias_n = vocab_size // self.mp if self.mp > 1 else vocab_size
        self.decoder_bias = nn.Parameter(torch.zeros(bias_n, dtype=dtype))
        if self.mp > 1:
            self.decoder_bias.is_mp = True
            self.decoder_bias.partition_dim = 0
        self.seq_relationship = nn.Linear(hidden_size, 2, dtype=dtype)

    def forward(self, x):
        pooled = self.pooler(x)
        h = self.layer_norm(self.activation(self.transform(x)))
        if self.mp > 1:
            from paddlefleetx_amd.parallel.tp import parallel_matmul
            scores = parallel_matmul(h, self.word_embeddings.weight,
                                     parallel_output=True) + self.decoder_bias
        else:
            scores = F.linear(h, self.word_embeddings.weight) \
                + self.decoder_bias
        return scores, self.seq_relationship(pooled)


class ErniePipeCriterion(nn.Module):
    """loss_fn(out, masked_lm_labels, next_sentence_labels) -> scalar;
    matches the PipelineModule loss contract (4-tuple batches)."""

    def __init__(self):
        super().__init__()
        self.inner = ErniePretrainingCriterion(with_nsp_loss=True)

    def forward(self, out, masked_lm_labels, next_sentence_labels):
        pred_scores, seq_rel = out
        mlm, nsp = self.inner(pred_scores, seq_rel, masked_lm_labels,
                              next_sentence_labels.long())
        return mlm + nsp


class ErnieForPretrainingPipe(PipelineModule):
    def __init__(self, vocab_size: int, hidden_size: int = 768,
                 num_hidden_layers: int = 12, num_attention_heads: int = 12,
                 intermediate_size: int = 3072, hidden_act: str = "gelu",
                 hidden_dropout_prob: float = 0.1,
                 attention_probs_dropout_prob: float = 0.1,
                 max_position_embeddings: int = 512, type_vocab_size: int = 2,
                 pad_token_id: int = 0, virtual_pp_degree: int = 1,
                 dtype: Optional[torch.dtype] = None, **unused: Any):
        descs = [
            SharedLayerDesc("ernie_embed", ErnieEmbeddingPipe,
                            shared_weight_attr="word_embeddings.weight",
                            vocab_size=vocab_size, hidden_size=hidden_size,
                            hidden_dropout_prob=hidden_dropout_prob,
                            max_position_embeddings=max_position_embeddings,
                            type_vocab_size=type_vocab_size,
                            pad_token_id=pad_token_id, dtype=dtype),
        ]
        for _ in range(num_hidden_layers):
            descs.append(LayerDesc(
                ErnieEncoderLayerPipe, hidden_size, num_attention_heads,
                intermediate_size, hidden_dropout=hidden_dropout_prob,
                attn_dropout=attention_probs_dropout_prob, dtype=dtype))
        descs.append(SharedLayerDesc(
            "ernie_embed", ErnieHeadsPipe,
            shared_weight_attr="word_embeddings.weight",
            vocab_size=vocab_size, hidden_size=hidden_size,
            hidden_act=hidden_act, dtype=dtype))
        super().__init__(descs, seg_method="layer:ErnieEncoderLayerPipe",
                         act_dtype=dtype or torch.float32,
                         num_virtual_stages=virtual_pp_degree)
        self.hidden_size = hidden_size
