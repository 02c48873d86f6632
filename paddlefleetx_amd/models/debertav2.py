"""DeBERTa-v2 encoder (disentangled attention).

Reference: ppfleetx/models/language_model/debertav2/modeling.py (1,323 LoC;
used as an Imagen text encoder alternative). Core behavior: content and
relative-position streams with content->position (c2p) and
position->content (p2c) disentangled attention terms, log-bucketed
relative distances, pre-LN=False (post-LN like BERT).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def make_log_bucket_position(relative_pos: torch.Tensor, bucket_size: int,
                             max_position: int) -> torch.Tensor:
    """Log-bucket mapping of relative distances (modeling.py
    make_log_bucket_position)."""
    sign = torch.sign(relative_pos)
    mid = bucket_size // 2
    abs_pos = torch.where((relative_pos < mid) & (relative_pos > -mid),
                          torch.full_like(relative_pos, mid - 1),
                          relative_pos.abs())
    log_pos = torch.ceil(
        torch.log(abs_pos.float() / mid) /
        math.log((max_position - 1) / mid) * (mid - 1)) + mid
    return torch.where(abs_pos <= mid, relative_pos,
                       (log_pos * sign).long())


def build_relative_position(q_len: int, k_len: int, bucket_size: int,
                            max_position: int, device) -> torch.Tensor:
    q_ids = torch.arange(q_len, device=device)
    k_ids = torch.arange(k_len, device=device)
    rel = q_ids[:, None] - k_ids[None, :]
    if bucket_size > 0 and max_position > 0:
        rel = make_log_bucket_position(rel, bucket_size, max_position)
    return rel.unsqueeze(0)  # [1, Q, K]


class DisentangledSelfAttention(nn.Module):
    def __init__(self, hidden_size: int, num_heads: int,
                 position_buckets: int = 256, max_relative_positions: int = 512,
                 attn_dropout: float = 0.1,
                 pos_att_type=("c2p", "p2c")):
        super().__init__()
        assert hidden_size % num_heads == 0
        self.num_heads = num_heads
        self.head_dim = hidden_size // num_heads
        self.qkv = nn.Linear(hidden_size, 3 * hidden_size)
        self.pos_att_type = tuple(pos_att_type)
        self.position_buckets = position_buckets
        self.max_relative_positions = max_relative_positions
        self.pos_ebd_size = position_buckets if position_buckets > 0 \
            else max_relative_positions
        if self.pos_att_type:
            self.pos_key_proj = nn.Linear(hidden_size, hidden_size)
            self.pos_query_proj = nn.Linear(hidden_size, hidden_size)
        self.dropout_p = attn_dropout
        # scale includes the number of attention score terms
        self.scale_factor = 1 + len(self.pos_att_type)

    def forward(self, x, rel_embeddings, attn_mask=None):
        B, S, C = x.shape
        h, d = self.num_heads, self.head_dim
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        q = q.view(B, S, h, d).transpose(1, 2)
        k = k.view(B, S, h, d).transpose(1, 2)
        v = v.view(B, S, h, d).transpose(1, 2)
        scale = 1.0 / math.sqrt(d * self.scale_factor)
        scores = torch.matmul(q, k.transpose(-1, -2)) * scale

        if self.pos_att_type:
            rel_pos = build_relative_position(S, S, self.position_buckets,
                                              self.max_relative_positions,
                                              x.device)  # [1, S, S]
            att_span = self.pos_ebd_size // 2
            # center slice of the relative embedding table
            mid = rel_embeddings.shape[0] // 2
            emb = rel_embeddings[mid - att_span:mid + att_span]  # [2a, C]
            if "c2p" in self.pos_att_type:
                pk = self.pos_key_proj(emb).view(-1, h, d).transpose(0, 1)
                c2p = torch.matmul(q, pk.transpose(-1, -2)) * scale
                idx = torch.clamp(rel_pos + att_span, 0, att_span * 2 - 1)
                c2p = torch.gather(
                    c2p, -1, idx.unsqueeze(1).expand(B, h, S, S))
                scores = scores + c2p
            if "p2c" in self.pos_att_type:
                pq = self.pos_query_proj(emb).view(-1, h, d).transpose(0, 1)
                p2c = torch.matmul(k, pq.transpose(-1, -2)) * scale
                idx = torch.clamp(-rel_pos + att_span, 0, att_span * 2 - 1)
                p2c = torch.gather(
                    p2c, -1, idx.unsqueeze(1).expand(B, h, S, S))
                scores = scores + p2c.transpose(-1, -2)

        if attn_mask is not None:
            scores = scores + attn_mask
        probs = scores.float().softmax(dim=-1).to(x.dtype)
        if self.dropout_p > 0 and self.training:
            probs = F.dropout(probs, self.dropout_p)
        o = torch.matmul(probs, v).transpose(1, 2).reshape(B, S, C)
        return o


class DebertaV2Layer(nn.Module):
    def __init__(self, hidden_size: int, num_heads: int,
                 intermediate_size: int, hidden_dropout: float = 0.1,
                 attn_dropout: float = 0.1, **attn_kw):
        super().__init__()
        self.attn = DisentangledSelfAttention(hidden_size, num_heads,
                                              attn_dropout=attn_dropout,
                                              **attn_kw)
        self.attn_out = nn.Linear(hidden_size, hidden_size)
        self.ln1 = nn.LayerNorm(hidden_size, eps=1e-7)
        self.fc1 = nn.Linear(hidden_size, intermediate_size)
        self.fc2 = nn.Linear(intermediate_size, hidden_size)
        self.ln2 = nn.LayerNorm(hidden_size, eps=1e-7)
        self.dropout_p = hidden_dropout

    def _drop(self, x):
        if self.dropout_p > 0 and self.training:
            return F.dropout(x, self.dropout_p)
        return x

    def forward(self, x, rel_embeddings, attn_mask=None):
        a = self.attn_out(self.attn(x, rel_embeddings, attn_mask))
        x = self.ln1(x + self._drop(a))
        h = self.fc2(F.gelu(self.fc1(x)))
        return self.ln2(x + self._drop(h))


class DebertaV2Model(nn.Module):
    def __init__(self, vocab_size: int = 128100, hidden_size: int = 768,
                 num_hidden_layers: int = 12, num_attention_heads: int = 12,
                 intermediate_size: int = 3072,
                 hidden_dropout_prob: float = 0.1,
                 attention_probs_dropout_prob: float = 0.1,
                 max_position_embeddings: int = 512,
                 position_buckets: int = 256,
                 max_relative_positions: int = -1,
                 pos_att_type=("c2p", "p2c"), pad_token_id: int = 0,
                 **unused):
        super().__init__()
        if max_relative_positions < 1:
            max_relative_positions = max_position_embeddings
        self.word_embeddings = nn.Embedding(vocab_size, hidden_size,
                                            padding_idx=pad_token_id)
        self.emb_ln = nn.LayerNorm(hidden_size, eps=1e-7)
        self.dropout_p = hidden_dropout_prob
        pos_ebd = position_buckets if position_buckets > 0 \
            else max_relative_positions
        self.rel_embeddings = nn.Embedding(pos_ebd * 2, hidden_size)
        self.rel_ln = nn.LayerNorm(hidden_size, eps=1e-7)
        self.layers = nn.ModuleList([
            DebertaV2Layer(hidden_size, num_attention_heads,
                           intermediate_size,
                           hidden_dropout=hidden_dropout_prob,
                           attn_dropout=attention_probs_dropout_prob,
                           position_buckets=position_buckets,
                           max_relative_positions=max_relative_positions,
                           pos_att_type=pos_att_type)
            for _ in range(num_hidden_layers)])

    def forward(self, input_ids, attention_mask=None):
        x = self.emb_ln(self.word_embeddings(input_ids))
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        add_mask = None
        if attention_mask is not None:
            add_mask = (1.0 - attention_mask[:, None, None, :].float()) * -1e4
        rel = self.rel_ln(self.rel_embeddings.weight)
        for layer in self.layers:
            x = layer(x, rel, add_mask)
        return x
