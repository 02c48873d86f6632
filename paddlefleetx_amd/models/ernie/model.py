"""ERNIE (BERT-style bidirectional encoder) with MLM+NSP pretraining heads.

Reference: ppfleetx/models/language_model/ernie/dygraph/single_model.py
  ErnieEmbeddings :34 (word+position+token_type(+task) embeddings, LN,
  dropout), ErniePooler :115, ErnieModel :131 (post-LN encoder),
  ErnieLMPredictionHead :401 (tied decoder weight), ErniePretrainingHeads
  :443, ErnieForPretraining :464, ErniePretrainingCriterion :591,
  ErnieForSequenceClassification :647; MoE wiring per the moe configs
  (driver config "ERNIE-MoE 64-expert EP8").

MI355X-native: fused LayerNorm + bias-gelu kernels; bidirectional flash
attention (gfx950) when there is no padding mask, additive-mask GEMM path
otherwise.

Tensor parallelism (reference ErnieModelHybrid, ernie/dygraph/
hybrid_model.py:167 + ernie/layers/distributed_transformer.py): when the
mp degree is >1 the QKV / out-proj / FFN projections become column/row
parallel over RCCL, the word embedding is vocab-sharded, and the tied MLM
decoder computes vocab-parallel logits consumed by ParallelCrossEntropy.
The per-rank qkv weight rows are laid out [q_shard; k_shard; v_shard] so
the single-card `view(B, S, 3, h_local, D)` split stays valid.
"""

from __future__ import annotations

import math
from typing import Any, Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.ops import FusedLayerNorm, bias_gelu
from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.parallel.tp import (ColumnParallelLinear,
                                          ParallelCrossEntropy,
                                          RowParallelLinear,
                                          VocabParallelEmbedding,
                                          parallel_matmul)


def _mp_degree() -> int:
    return get_hcg().get_model_parallel_world_size()


class ErnieEmbeddings(nn.Module):
    def __init__(self, vocab_size: int, hidden_size: int = 768,
                 hidden_dropout_prob: float = 0.1,
                 max_position_embeddings: int = 512,
                 type_vocab_size: int = 2, pad_token_id: int = 0,
                 task_type_vocab_size: int = 3, task_id: int = 0,
                 use_task_id: bool = False,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        if _mp_degree() > 1:
            # vocab-sharded table; tied MLM decoder uses the same shard
            self.word_embeddings = VocabParallelEmbedding(
                vocab_size, hidden_size, dtype=dtype)
        else:
            self.word_embeddings = nn.Embedding(vocab_size, hidden_size,
                                                padding_idx=pad_token_id,
                                                dtype=dtype)
        self.position_embeddings = nn.Embedding(max_position_embeddings,
                                                hidden_size, dtype=dtype)
        self.type_vocab_size = type_vocab_size
        if type_vocab_size > 0:
            self.token_type_embeddings = nn.Embedding(type_vocab_size,
                                                      hidden_size, dtype=dtype)
        self.use_task_id = use_task_id
        self.task_id = task_id
        if use_task_id:
            self.task_type_embeddings = nn.Embedding(task_type_vocab_size,
                                                     hidden_size, dtype=dtype)
        self.layer_norm = FusedLayerNorm(hidden_size, eps=1e-12, dtype=dtype)
        self.dropout_p = hidden_dropout_prob

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                task_type_ids=None):
        B, S = input_ids.shape
        x = self.word_embeddings(input_ids)
        if position_ids is None:
            position_ids = torch.arange(S, device=input_ids.device) \
                .unsqueeze(0).expand(B, S)
        x = x + self.position_embeddings(position_ids)
        if self.type_vocab_size > 0:
            if token_type_ids is None:
                token_type_ids = torch.zeros_like(input_ids)
            x = x + self.token_type_embeddings(token_type_ids)
        if self.use_task_id:
            if task_type_ids is None:
                task_type_ids = torch.full_like(input_ids, self.task_id)
            x = x + self.task_type_embeddings(task_type_ids)
        x = self.layer_norm(x)
        if self.dropout_p > 0.0 and self.training:
            x = F.dropout(x, p=self.dropout_p)
        return x


class ErnieSelfAttention(nn.Module):
    """Bidirectional MHA with optional additive padding mask."""

    def __init__(self, hidden_size: int, num_heads: int,
                 attn_dropout: float = 0.1,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        assert hidden_size % num_heads == 0
        mp = _mp_degree()
        assert num_heads % mp == 0
        self.num_heads = num_heads
        self.nh_local = num_heads // mp
        self.head_dim = hidden_size // num_heads
        self.scale = 1.0 / math.sqrt(self.head_dim)
        if mp > 1:
            # per-rank rows are [q_shard; k_shard; v_shard]
            self.qkv = ColumnParallelLinear(hidden_size, 3 * hidden_size,
                                            bias=True, dtype=dtype)
            self.out_proj = RowParallelLinear(hidden_size, hidden_size,
                                              bias=True, dtype=dtype)
        else:
            self.qkv = nn.Linear(hidden_size, 3 * hidden_size, dtype=dtype)
            self.out_proj = nn.Linear(hidden_size, hidden_size, dtype=dtype)
        self.attn_dropout_p = attn_dropout

    def forward(self, x, attn_mask=None):
        B, S, C = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.nh_local, self.head_dim)
        q, k, v = qkv.unbind(dim=2)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        use_kernel = (x.is_cuda and x.dtype == torch.bfloat16
                      and attn_mask is None and self.head_dim in (64, 128)
                      and not (self.attn_dropout_p > 0.0 and self.training))
        if use_kernel:
            from paddlefleetx_amd.ops import flash_attention
            o = flash_attention(q, k, v, causal=False, scale=self.scale)
        else:
            scores = torch.matmul(q, k.transpose(-1, -2)) * self.scale
            if attn_mask is not None:
                scores = scores + attn_mask
            probs = F.softmax(scores.float(), dim=-1).to(x.dtype)
            if self.attn_dropout_p > 0.0 and self.training:
                probs = F.dropout(probs, p=self.attn_dropout_p)
            o = torch.matmul(probs, v)
        o = o.transpose(1, 2).reshape(B, S, -1)  # [B, S, h_local*D]
        return self.out_proj(o)


class ErnieEncoderLayer(nn.Module):
    """Post-LN encoder layer (paddle TransformerEncoderLayer default,
    normalize_before=False). FFN is replaceable by a MoE expert module."""

    def __init__(self, hidden_size: int, num_heads: int,
                 intermediate_size: int, hidden_dropout: float = 0.1,
                 attn_dropout: float = 0.1,
                 expert_module: Optional[nn.Module] = None,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.attn = ErnieSelfAttention(hidden_size, num_heads,
                                       attn_dropout=attn_dropout, dtype=dtype)
        self.ln1 = FusedLayerNorm(hidden_size, eps=1e-12, dtype=dtype)
        self.ln2 = FusedLayerNorm(hidden_size, eps=1e-12, dtype=dtype)
        if expert_module is not None:
            self.ffn = expert_module
        elif _mp_degree() > 1:
            mp = _mp_degree()
            self.fc1 = ColumnParallelLinear(hidden_size, intermediate_size,
                                            bias=False, dtype=dtype)
            self.fc1_bias = nn.Parameter(
                torch.zeros(intermediate_size // mp, dtype=dtype))
            self.fc1_bias.is_mp = True
            self.fc1_bias.partition_dim = 0
            self.fc2 = RowParallelLinear(intermediate_size, hidden_size,
                                         bias=True, dtype=dtype)
            self.ffn = None
        else:
            self.fc1 = nn.Linear(hidden_size, intermediate_size, bias=False,
                                 dtype=dtype)
            self.fc1_bias = nn.Parameter(
                torch.zeros(intermediate_size, dtype=dtype))
            self.fc2 = nn.Linear(intermediate_size, hidden_size, dtype=dtype)
            self.ffn = None
        self.dropout_p = hidden_dropout

    def _drop(self, x):
        if self.dropout_p > 0.0 and self.training:
            return F.dropout(x, p=self.dropout_p)
        return x

    def forward(self, x, attn_mask=None):
        x = self.ln1(x + self._drop(self.attn(x, attn_mask)))
        if self.ffn is not None:
            h = self.ffn(x)
        else:
            h = self.fc2(bias_gelu(self.fc1(x), self.fc1_bias))
        return self.ln2(x + self._drop(h))


class ErniePooler(nn.Module):
    def __init__(self, hidden_size: int, dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.dense = nn.Linear(hidden_size, hidden_size, dtype=dtype)

    def forward(self, hidden_states):
        return torch.tanh(self.dense(hidden_states[:, 0]))


class ErnieModel(nn.Module):
    def __init__(self, vocab_size: int, hidden_size: int = 768,
                 num_hidden_layers: int = 12, num_attention_heads: int = 12,
                 intermediate_size: int = 3072, hidden_act: str = "gelu",
                 hidden_dropout_prob: float = 0.1,
                 attention_probs_dropout_prob: float = 0.1,
                 max_position_embeddings: int = 512, type_vocab_size: int = 2,
                 initializer_range: float = 0.02, pad_token_id: int = 0,
                 task_type_vocab_size: int = 3, task_id: int = 0,
                 use_task_id: bool = False,
                 moe_configs: Optional[Dict[str, Any]] = None,
                 dtype: Optional[torch.dtype] = None, **unused: Any):
        super().__init__()
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.hidden_act = hidden_act
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.embeddings = ErnieEmbeddings(
            vocab_size, hidden_size, hidden_dropout_prob,
            max_position_embeddings, type_vocab_size, pad_token_id,
            task_type_vocab_size, task_id, use_task_id, dtype=dtype)

        def _expert():
            if not (moe_configs and moe_configs.get("expert_mode", False)):
                return None
            from paddlefleetx_amd.models.moe import MoELayer
            return MoELayer(hidden_size, intermediate_size,
                            num_experts=moe_configs.get("num_experts", 1),
                            gate=moe_configs.get("gate", "gshard"),
                            top_k=moe_configs.get("top_k", 2),
                            capacity_factor=moe_configs.get("capacity_factor"),
                            dtype=dtype)

        self.encoder = nn.ModuleList([
            ErnieEncoderLayer(hidden_size, num_attention_heads,
                              intermediate_size,
                              hidden_dropout=hidden_dropout_prob,
                              attn_dropout=attention_probs_dropout_prob,
                              expert_module=_expert(), dtype=dtype)
            for _ in range(num_hidden_layers)])
        self.pooler = ErniePooler(hidden_size, dtype=dtype)
        self.apply(self._init_weights)

    def _init_weights(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.trunc_normal_(m.weight, std=self.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None):
        """attention_mask: [B, S] with 1 for valid tokens, or None."""
        add_mask = None
        if attention_mask is not None:
            # additive mask [B, 1, 1, S] (single_model.py builds -1e4 * (1-m))
            add_mask = (1.0 - attention_mask[:, None, None, :].float()) * -1e4
        x = self.embeddings(input_ids, token_type_ids, position_ids)
        for layer in self.encoder:
            x = layer(x, add_mask)
        return x, self.pooler(x)


class ErnieLMPredictionHead(nn.Module):
    """MLM head; decoder weight tied to the word embedding."""

    def __init__(self, hidden_size: int, vocab_size: int,
                 activation: str = "gelu",
                 embedding_weights: Optional[torch.Tensor] = None,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.transform = nn.Linear(hidden_size, hidden_size, dtype=dtype)
        self.activation = getattr(F, activation)
        self.layer_norm = FusedLayerNorm(hidden_size, eps=1e-12, dtype=dtype)
        self.mp = _mp_degree()
        if embedding_weights is not None:
            self.decoder_weight = embedding_weights
        else:
            # untied: shard the decoder weight on vocab like the tied
            # VocabParallelEmbedding so the parallel_matmul output and the
            # sharded bias agree at [.., V/mp]
            vocab_n = vocab_size // self.mp if self.mp > 1 else vocab_size
            self.decoder_weight = nn.Parameter(
                torch.empty(vocab_n, hidden_size, dtype=dtype))
            nn.init.normal_(self.decoder_weight, std=0.02)
            if self.mp > 1:
                self.decoder_weight.is_mp = True
                self.decoder_weight.partition_dim = 0
        bias_n = vocab_size // self.mp if self.mp > 1 else vocab_size
        self.decoder_bias = nn.Parameter(torch.zeros(bias_n, dtype=dtype))
        if self.mp > 1:
            self.decoder_bias.is_mp = True
            self.decoder_bias.partition_dim = 0

    def forward(self, hidden_states, masked_positions=None):
        if masked_positions is not None:
            hs = hidden_states.reshape(-1, hidden_states.shape[-1])
            hidden_states = hs.index_select(0, masked_positions)
        h = self.layer_norm(self.activation(self.transform(hidden_states)))
        if self.mp > 1:
            # vocab-parallel logits [.., V/mp] for ParallelCrossEntropy
            return parallel_matmul(h, self.decoder_weight,
                                   parallel_output=True) + self.decoder_bias
        return F.linear(h, self.decoder_weight) + self.decoder_bias


class ErniePretrainingHeads(nn.Module):
    def __init__(self, hidden_size: int, vocab_size: int,
                 activation: str = "gelu",
                 embedding_weights: Optional[torch.Tensor] = None,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.predictions = ErnieLMPredictionHead(hidden_size, vocab_size,
                                                 activation,
                                                 embedding_weights, dtype)
        self.seq_relationship = nn.Linear(hidden_size, 2, dtype=dtype)

    def forward(self, sequence_output, pooled_output, masked_positions=None):
        return (self.predictions(sequence_output, masked_positions),
                self.seq_relationship(pooled_output))


class ErnieForPretraining(nn.Module):
    def __init__(self, ernie: ErnieModel):
        super().__init__()
        self.ernie = ernie
        dtype = ernie.embeddings.word_embeddings.weight.dtype
        self.cls = ErniePretrainingHeads(
            ernie.hidden_size, ernie.vocab_size, ernie.hidden_act,
            embedding_weights=ernie.embeddings.word_embeddings.weight,
            dtype=dtype)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, masked_positions=None):
        seq_out, pooled = self.ernie(input_ids, token_type_ids, position_ids,
                                     attention_mask)
        return self.cls(seq_out, pooled, masked_positions)


class ErniePretrainingCriterion(nn.Module):
    """MLM CE (ignore_index=-1) + NSP CE (single_model.py:591-646)."""

    def __init__(self, with_nsp_loss: bool = True):
        super().__init__()
        self.with_nsp_loss = with_nsp_loss
        self.parallel_ce = ParallelCrossEntropy(ignore_index=-1) \
            if _mp_degree() > 1 else None

    def forward(self, prediction_scores, seq_relationship_score,
                masked_lm_labels, next_sentence_labels=None):
        if self.parallel_ce is not None:
            flat_labels = masked_lm_labels.reshape(-1)
            loss_vec = self.parallel_ce(
                prediction_scores.reshape(-1, prediction_scores.shape[-1]),
                flat_labels)
            nvalid = (flat_labels != -1).sum().clamp(min=1)
            mlm = loss_vec.float().sum() / nvalid.float()
        else:
            mlm = F.cross_entropy(
                prediction_scores.float().reshape(
                    -1, prediction_scores.shape[-1]),
                masked_lm_labels.reshape(-1), ignore_index=-1)
        if not self.with_nsp_loss:
            return mlm
        nsp = F.cross_entropy(seq_relationship_score.float().reshape(-1, 2),
                              next_sentence_labels.reshape(-1))
        return mlm, nsp


class ErnieForSequenceClassification(nn.Module):
    def __init__(self, ernie: ErnieModel, num_classes: int = 2,
                 dropout: Optional[float] = None):
        super().__init__()
        self.ernie = ernie
        self.dropout_p = dropout if dropout is not None else 0.1
        self.classifier = nn.Linear(
            ernie.hidden_size, num_classes,
            dtype=ernie.embeddings.word_embeddings.weight.dtype)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None):
        _, pooled = self.ernie(input_ids, token_type_ids, position_ids,
                               attention_mask)
        if self.dropout_p > 0.0 and self.training:
            pooled = F.dropout(pooled, p=self.dropout_p)
        return self.classifier(pooled)
