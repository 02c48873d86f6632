"""T5 encoder/decoder (relative-position-bias transformer).

Reference: ppfleetx/models/language_model/t5/modeling.py — T5LayerNorm
:473 (no-mean RMS norm -> our gfx950 FusedRMSNorm), T5DenseActDense :504,
T5DenseGatedActDense :520, T5Attention with bucketed relative position
bias :559, T5Block :890, T5Stack :1033, T5EncoderModel :1318 (the Imagen
text encoder). Implemented from the architecture, sized by T5Config
kwargs.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.ops import FusedRMSNorm

__all__ = ["T5Config", "T5Attention", "T5Block", "T5Stack",
           "T5EncoderModel", "T5Model"]


class T5Config:
    def __init__(self, vocab_size=32128, d_model=512, d_kv=64, d_ff=2048,
                 num_layers=6, num_decoder_layers=None, num_heads=8,
                 relative_attention_num_buckets=32,
                 relative_attention_max_distance=128, dropout_rate=0.1,
                 layer_norm_epsilon=1e-6, feed_forward_proj="relu",
                 is_gated_act=None, **unused):
        self.vocab_size = vocab_size
        self.d_model = d_model
        self.d_kv = d_kv
        self.d_ff = d_ff
        self.num_layers = num_layers
        self.num_decoder_layers = num_decoder_layers or num_layers
        self.num_heads = num_heads
        self.relative_attention_num_buckets = relative_attention_num_buckets
        self.relative_attention_max_distance = relative_attention_max_distance
        self.dropout_rate = dropout_rate
        self.layer_norm_epsilon = layer_norm_epsilon
        acts = feed_forward_proj.split("-")
        self.dense_act_fn = acts[-1]
        self.is_gated_act = (acts[0] == "gated") if is_gated_act is None \
            else is_gated_act


def _act(name: str):
    return {"relu": F.relu, "gelu": F.gelu,
            "gelu_new": lambda x: F.gelu(x, approximate="tanh"),
            "silu": F.silu}[name]


class T5DenseActDense(nn.Module):
    def __init__(self, cfg: T5Config):
        super().__init__()
        self.wi = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        self.wo = nn.Linear(cfg.d_ff, cfg.d_model, bias=False)
        self.act = _act(cfg.dense_act_fn)
        self.dropout_p = cfg.dropout_rate

    def forward(self, x):
        x = self.act(self.wi(x))
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        return self.wo(x)


class T5DenseGatedActDense(nn.Module):
    def __init__(self, cfg: T5Config):
        super().__init__()
        self.wi_0 = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        self.wi_1 = nn.Linear(cfg.d_model, cfg.d_ff, bias=False)
        self.wo = nn.Linear(cfg.d_ff, cfg.d_model, bias=False)
        self.act = _act(cfg.dense_act_fn)
        self.dropout_p = cfg.dropout_rate

    def forward(self, x):
        x = self.act(self.wi_0(x)) * self.wi_1(x)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        return self.wo(x)


class T5LayerFF(nn.Module):
    def __init__(self, cfg: T5Config):
        super().__init__()
        self.DenseReluDense = T5DenseGatedActDense(cfg) if cfg.is_gated_act \
            else T5DenseActDense(cfg)
        self.layer_norm = FusedRMSNorm(cfg.d_model, eps=cfg.layer_norm_epsilon)
        self.dropout_p = cfg.dropout_rate

    def forward(self, x):
        h = self.DenseReluDense(self.layer_norm(x))
        if self.dropout_p > 0 and self.training:
            h = F.dropout(h, self.dropout_p)
        return x + h


def relative_position_bucket(relative_position, bidirectional: bool,
                             num_buckets: int, max_distance: int):
    """T5 bucketing (modeling.py T5Attention._relative_position_bucket)."""
    ret = torch.zeros_like(relative_position)
    n = -relative_position
    if bidirectional:
        num_buckets //= 2
        ret = ret + (n < 0).long() * num_buckets
        n = n.abs()
    else:
        n = torch.clamp(n, min=0)
    max_exact = num_buckets // 2
    is_small = n < max_exact
    val_large = max_exact + (
        torch.log(n.float() / max_exact + 1e-6) /
        math.log(max_distance / max_exact) * (num_buckets - max_exact)).long()
    val_large = torch.clamp(val_large, max=num_buckets - 1)
    return ret + torch.where(is_small, n, val_large)


class T5Attention(nn.Module):
    def __init__(self, cfg: T5Config, has_relative_bias: bool = False,
                 is_decoder: bool = False):
        super().__init__()
        self.is_decoder = is_decoder
        self.has_relative_bias = has_relative_bias
        self.d_model = cfg.d_model
        self.n_heads = cfg.num_heads
        self.d_kv = cfg.d_kv
        inner = self.n_heads * self.d_kv
        self.q = nn.Linear(cfg.d_model, inner, bias=False)
        self.k = nn.Linear(cfg.d_model, inner, bias=False)
        self.v = nn.Linear(cfg.d_model, inner, bias=False)
        self.o = nn.Linear(inner, cfg.d_model, bias=False)
        self.num_buckets = cfg.relative_attention_num_buckets
        self.max_distance = cfg.relative_attention_max_distance
        if has_relative_bias:
            self.relative_attention_bias = nn.Embedding(self.num_buckets,
                                                        self.n_heads)
        self.dropout_p = cfg.dropout_rate

    def compute_bias(self, q_len: int, k_len: int, device) -> torch.Tensor:
        ctx = torch.arange(q_len, device=device)[:, None]
        mem = torch.arange(k_len, device=device)[None, :]
        buckets = relative_position_bucket(mem - ctx,
                                           bidirectional=not self.is_decoder,
                                           num_buckets=self.num_buckets,
                                           max_distance=self.max_distance)
        bias = self.relative_attention_bias(buckets)  # [q, k, h]
        return bias.permute(2, 0, 1).unsqueeze(0)     # [1, h, q, k]

    def forward(self, x, kv=None, mask=None, position_bias=None):
        B, S, _ = x.shape
        kv = x if kv is None else kv
        Sk = kv.shape[1]
        q = self.q(x).view(B, S, self.n_heads, self.d_kv).transpose(1, 2)
        k = self.k(kv).view(B, Sk, self.n_heads, self.d_kv).transpose(1, 2)
        v = self.v(kv).view(B, Sk, self.n_heads, self.d_kv).transpose(1, 2)
        # T5 omits the 1/sqrt(d) scale (folded into init)
        scores = torch.matmul(q, k.transpose(-1, -2))
        if position_bias is None and self.has_relative_bias:
            position_bias = self.compute_bias(S, Sk, x.device)
        if position_bias is not None:
            scores = scores + position_bias
        if mask is not None:
            scores = scores + mask
        probs = F.softmax(scores.float(), dim=-1).to(x.dtype)
        if self.dropout_p > 0 and self.training:
            probs = F.dropout(probs, self.dropout_p)
        o = torch.matmul(probs, v).transpose(1, 2).reshape(B, S, -1)
        return self.o(o), position_bias


class T5Block(nn.Module):
    def __init__(self, cfg: T5Config, has_relative_bias: bool = False,
                 is_decoder: bool = False):
        super().__init__()
        self.is_decoder = is_decoder
        self.self_attn = T5Attention(cfg, has_relative_bias, is_decoder)
        self.self_ln = FusedRMSNorm(cfg.d_model, eps=cfg.layer_norm_epsilon)
        if is_decoder:
            self.cross_attn = T5Attention(cfg, False, is_decoder)
            self.cross_ln = FusedRMSNorm(cfg.d_model,
                                         eps=cfg.layer_norm_epsilon)
        self.ff = T5LayerFF(cfg)
        self.dropout_p = cfg.dropout_rate

    def _drop(self, x):
        if self.dropout_p > 0 and self.training:
            return F.dropout(x, self.dropout_p)
        return x

    def forward(self, x, enc_out=None, self_mask=None, cross_mask=None,
                position_bias=None):
        a, position_bias = self.self_attn(self.self_ln(x), mask=self_mask,
                                          position_bias=position_bias)
        x = x + self._drop(a)
        if self.is_decoder and enc_out is not None:
            c, _ = self.cross_attn(self.cross_ln(x), kv=enc_out,
                                   mask=cross_mask)
            x = x + self._drop(c)
        return self.ff(x), position_bias


class T5Stack(nn.Module):
    def __init__(self, cfg: T5Config, embed: nn.Embedding,
                 is_decoder: bool = False):
        super().__init__()
        self.embed_tokens = embed
        self.is_decoder = is_decoder
        n = cfg.num_decoder_layers if is_decoder else cfg.num_layers
        self.block = nn.ModuleList([
            T5Block(cfg, has_relative_bias=(i == 0), is_decoder=is_decoder)
            for i in range(n)])
        self.final_layer_norm = FusedRMSNorm(cfg.d_model,
                                             eps=cfg.layer_norm_epsilon)
        self.dropout_p = cfg.dropout_rate

    def forward(self, input_ids, enc_out=None, attention_mask=None,
                enc_attention_mask=None):
        x = self.embed_tokens(input_ids)
        B, S = input_ids.shape
        self_mask = None
        if self.is_decoder:
            causal = torch.ones(S, S, device=x.device).tril()
            self_mask = (1.0 - causal)[None, None] * -1e9
            if attention_mask is not None:
                self_mask = self_mask + \
                    (1.0 - attention_mask[:, None, None, :].float()) * -1e9
        elif attention_mask is not None:
            self_mask = (1.0 - attention_mask[:, None, None, :].float()) * -1e9
        cross_mask = None
        if enc_attention_mask is not None:
            cross_mask = (1.0 - enc_attention_mask[:, None, None, :].float()) \
                * -1e9
        pos_bias = None
        for blk in self.block:
            x, pos_bias = blk(x, enc_out=enc_out, self_mask=self_mask,
                              cross_mask=cross_mask, position_bias=pos_bias)
        return self.final_layer_norm(x)


class T5EncoderModel(nn.Module):
    """Text encoder (Imagen conditioning; modeling.py:1318)."""

    def __init__(self, **kwargs):
        super().__init__()
        cfg = T5Config(**kwargs)
        self.config = cfg
        self.shared = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.encoder = T5Stack(cfg, self.shared, is_decoder=False)

    def forward(self, input_ids, attention_mask=None):
        return self.encoder(input_ids, attention_mask=attention_mask)


class T5Model(nn.Module):
    def __init__(self, **kwargs):
        super().__init__()
        cfg = T5Config(**kwargs)
        self.config = cfg
        self.shared = nn.Embedding(cfg.vocab_size, cfg.d_model)
        self.encoder = T5Stack(cfg, self.shared, is_decoder=False)
        self.decoder = T5Stack(cfg, self.shared, is_decoder=True)
        self.lm_head = nn.Linear(cfg.d_model, cfg.vocab_size, bias=False)

    def forward(self, input_ids, decoder_input_ids, attention_mask=None,
                decoder_attention_mask=None):
        enc = self.encoder(input_ids, attention_mask=attention_mask)
        dec = self.decoder(decoder_input_ids, enc_out=enc,
                           attention_mask=decoder_attention_mask,
                           enc_attention_mask=attention_mask)
        return self.lm_head(dec * (self.config.d_model ** -0.5))


class T5ForConditionalGeneration(nn.Module):
    """Seq2seq head with teacher-forced loss + greedy decode
    (reference t5/modeling.py T5ForConditionalGeneration surface)."""

    def __init__(self, decoder_start_token_id: int = 0, eos_token_id: int = 1,
                 **kwargs):
        super().__init__()
        self.model = T5Model(**kwargs)
        self.decoder_start_token_id = decoder_start_token_id
        self.eos_token_id = eos_token_id

    @property
    def config(self):
        return self.model.config

    def _shift_right(self, labels):
        start = torch.full_like(labels[:, :1], self.decoder_start_token_id)
        return torch.cat([start, labels[:, :-1]], dim=1)

    def forward(self, input_ids, labels=None, decoder_input_ids=None,
                attention_mask=None):
        if decoder_input_ids is None:
            assert labels is not None
            decoder_input_ids = self._shift_right(labels)
        logits = self.model(input_ids, decoder_input_ids,
                            attention_mask=attention_mask)
        if labels is None:
            return logits
        loss = F.cross_entropy(logits.float().flatten(0, 1),
                               labels.flatten(), ignore_index=-100)
        return loss, logits

    @torch.no_grad()
    def generate(self, input_ids, max_length: int = 20,
                 attention_mask=None):
        """Greedy decode (the reference's default strategy)."""
        enc = self.model.encoder(input_ids, attention_mask=attention_mask)
        B = input_ids.shape[0]
        dec = torch.full((B, 1), self.decoder_start_token_id,
                         dtype=torch.long, device=input_ids.device)
        finished = torch.zeros(B, dtype=torch.bool, device=input_ids.device)
        for _ in range(max_length):
            out = self.model.decoder(dec, enc_out=enc,
                                     enc_attention_mask=attention_mask)
            logits = self.model.lm_head(
                out[:, -1] * (self.config.d_model ** -0.5))
            nxt = logits.argmax(-1, keepdim=True)
            nxt = torch.where(finished.unsqueeze(1),
                              torch.full_like(nxt, self.eos_token_id), nxt)
            dec = torch.cat([dec, nxt], dim=1)
            finished |= nxt.squeeze(1) == self.eos_token_id
            if bool(finished.all()):
                break
        return dec
