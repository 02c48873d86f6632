"""GPT-2/3 decoder, MI355X-native.

One implementation covers the reference's single-card GPTModel
(ppfleetx/models/language_model/gpt/dygraph/single_model.py:608) and the
tensor-parallel GPTModelHybrid (hybrid_model.py:739): the parallel layers
degenerate at mp==1. Hot path: fused-QKV column-parallel GEMM ->
hand-written gfx950 flash attention -> row-parallel GEMM -> fused
bias-gelu FFN -> fused LayerNorm (all in paddlefleetx_amd.ops).

Sequence-parallel (hybrid_model.py:727-735, sequence_parallel_utils.py) is
handled by the sp variants of the linears when `sequence_parallel=True`
(activations [s/mp, b, h] between layers).
"""

from __future__ import annotations

import math
from typing import Any, Dict, List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint

from paddlefleetx_amd.ops import (FusedLayerNorm, bias_gelu, flash_attention,
                                  flash_attention_packed,
                                  fused_softmax_causal)
from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.parallel.rng import (checkpoint_rng_context,
                                            model_parallel_rng)
from paddlefleetx_amd.parallel.tp import (ColumnParallelLinear,
                                          ParallelCrossEntropy,
                                          RowParallelLinear,
                                          VocabParallelEmbedding,
                                          parallel_matmul)
from paddlefleetx_amd.parallel import sp as sp_ops

__all__ = [
    "GPTModel", "GPTForPretraining", "GPTPretrainingCriterion",
    "MultiHeadAttention", "TransformerDecoderLayer", "GPTEmbeddings",
]

KVCache = Tuple[torch.Tensor, torch.Tensor]  # ([B,H,S,D] k, [B,H,S,D] v)


class MultiHeadAttention(nn.Module):
    """Fused-QKV column-parallel attention (hybrid_model.py:96-392).

    fused_attn=True -> gfx950 flash-attention kernel (O(S) memory);
    otherwise explicit QK^T + fused causal softmax + PV (core_attn,
    hybrid_model.py:303-346) which supports attention dropout.
    """

    def __init__(self, hidden_size: int, num_heads: int,
                 attn_dropout: float = 0.0, fused_attn: bool = True,
                 dtype: Optional[torch.dtype] = None,
                 sequence_parallel: bool = False,
                 init_std: float = 0.02, output_layer_init_std: float = 0.02,
                 cp_backend: str = "ulysses", cp_zigzag: bool = False):
        super().__init__()
        assert cp_backend in ("ulysses", "ring")
        self.cp_backend = cp_backend
        self.cp_zigzag = cp_zigzag
        mp = get_hcg().get_model_parallel_world_size()
        assert num_heads % mp == 0, f"heads {num_heads} not divisible by mp {mp}"
        assert hidden_size % num_heads == 0
        self.num_heads = num_heads
        self.num_heads_local = num_heads // mp
        self.head_dim = hidden_size // num_heads
        self.hidden_size = hidden_size
        self.attn_dropout_p = attn_dropout
        self.fused_attn = fused_attn
        self.sequence_parallel = sequence_parallel
        if sequence_parallel:
            self.qkv = sp_ops.ColumnSequenceParallelLinear(
                hidden_size, 3 * hidden_size, bias=True, dtype=dtype,
                init_std=init_std)
            self.out_proj = sp_ops.RowSequenceParallelLinear(
                hidden_size, hidden_size, bias=True, dtype=dtype,
                init_std=output_layer_init_std)
        else:
            self.qkv = ColumnParallelLinear(hidden_size, 3 * hidden_size,
                                            bias=True, dtype=dtype,
                                            init_std=init_std)
            self.out_proj = RowParallelLinear(hidden_size, hidden_size,
                                              bias=True, dtype=dtype,
                                              init_std=output_layer_init_std)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, x: torch.Tensor, cache: Optional[KVCache] = None,
                use_cache: bool = False
                ) -> Tuple[torch.Tensor, Optional[KVCache]]:
        qkv = self.qkv(x)
        cp = get_hcg().get_context_parallel_world_size()
        if (cp > 1 and cache is None and not use_cache
                and not self.sequence_parallel):
            # Ulysses CP: x is the [B, S/cp, H] sequence shard; all-to-all
            # inside attention trades seq for heads (parallel/cp.py)
            assert self.attn_dropout_p == 0.0, \
                "attention dropout unsupported under context parallel"
            if self.cp_backend != "ring":
                # ring keeps all heads local; only the a2a path shards them
                assert self.num_heads_local % cp == 0, \
                    f"heads/mp {self.num_heads_local} not divisible by cp {cp}"
            B, Sl, _ = qkv.shape
            qkv = qkv.view(B, Sl, self.num_heads_local, 3 * self.head_dim)
            q, k, v = qkv.split(self.head_dim, dim=-1)  # [B, S/cp, h, D]
            if self.cp_backend == "ring":
                from paddlefleetx_amd.parallel.ring import RingAttention
                ca = RingAttention(scale=self.scale, causal=True,
                                   zigzag=self.cp_zigzag)
            else:
                from paddlefleetx_amd.parallel.cp import UlyssesAttention
                ca = UlyssesAttention(scale=self.scale, causal=True)
            o = ca(q, k, v).reshape(B, Sl, -1)
            return self.out_proj(o), None
        p_att = self.attn_dropout_p if self.training else 0.0
        if (self.fused_attn and cache is None and not use_cache
                and not self.sequence_parallel
                and (p_att == 0.0 or qkv.is_cuda)):
            # packed fast path: kernel reads the fused-QKV linear output
            # directly and writes [B, S, h*D] — zero layout copies.
            # Attention dropout runs IN the kernel (Philox mask keyed by
            # the per-rank local seed via model_parallel_rng — reference
            # hybrid_model.py:328 RNG-tracker dropout)
            B, S, _ = qkv.shape
            packed = qkv.view(B, S, self.num_heads_local, 3, self.head_dim)
            if p_att > 0.0:
                # mp-rank-local Philox seed for the in-kernel mask; the
                # context seeds the generator, so enter it ONLY when a
                # seed is drawn (hipGraph capture forbids reseeding)
                with model_parallel_rng():
                    o = flash_attention_packed(packed, self.num_heads_local,
                                               scale=self.scale,
                                               p_drop=p_att)
            else:
                o = flash_attention_packed(packed, self.num_heads_local,
                                           scale=self.scale)
            return self.out_proj(o), None
        if self.sequence_parallel:
            # x: [s/mp, B, H]; qkv allgathered the seq dim -> [S, B, 3H/mp]
            S, B = qkv.shape[0], qkv.shape[1]
            qkv = qkv.view(S, B, self.num_heads_local, 3 * self.head_dim)
            q, k, v = qkv.split(self.head_dim, dim=-1)
            q = q.permute(1, 2, 0, 3)  # [B, h, S, D]
            k = k.permute(1, 2, 0, 3)
            v = v.permute(1, 2, 0, 3)
        else:
            # x: [B, S, H]
            B, S, _ = x.shape
            qkv = qkv.view(B, S, self.num_heads_local, 3 * self.head_dim)
            q, k, v = qkv.split(self.head_dim, dim=-1)
            q = q.transpose(1, 2)  # [B, h, S, D]
            k = k.transpose(1, 2)
            v = v.transpose(1, 2)

        if cache is not None:
            k = torch.cat([cache[0], k], dim=2)
            v = torch.cat([cache[1], v], dim=2)
        new_cache = (k, v) if use_cache else None

        if self.fused_attn and not (cache is not None and S == 1) and (
                p_att == 0.0 or q.is_cuda):
            if p_att > 0.0:
                with model_parallel_rng():
                    o = flash_attention(q, k, v, causal=True,
                                        scale=self.scale, p_drop=p_att)
            else:
                o = flash_attention(q, k, v, causal=True, scale=self.scale)
        else:
            scores = torch.matmul(q, k.transpose(-1, -2))
            probs = fused_softmax_causal(scores, self.scale)
            if self.attn_dropout_p > 0.0 and self.training:
                with model_parallel_rng():
                    probs = F.dropout(probs, p=self.attn_dropout_p)
            o = torch.matmul(probs, v)
        if self.sequence_parallel:
            o = o.permute(2, 0, 1, 3).reshape(S, B, -1)  # [S, B, H/mp]
        else:
            o = o.transpose(1, 2).reshape(B, S, -1)  # [B, S, H/mp]
        out = self.out_proj(o)
        return out, new_cache


class FFN(nn.Module):
    """Column-parallel up-proj -> fused bias-gelu -> row-parallel down-proj."""

    def __init__(self, hidden_size: int, ffn_hidden_size: int,
                 dtype: Optional[torch.dtype] = None, init_std: float = 0.02,
                 output_layer_init_std: float = 0.02,
                 sequence_parallel: bool = False):
        super().__init__()
        if sequence_parallel:
            self.up = sp_ops.ColumnSequenceParallelLinear(
                hidden_size, ffn_hidden_size, bias=False, dtype=dtype,
                init_std=init_std)
            self.down = sp_ops.RowSequenceParallelLinear(
                ffn_hidden_size, hidden_size, bias=True, dtype=dtype,
                init_std=output_layer_init_std)
        else:
            self.up = ColumnParallelLinear(hidden_size, ffn_hidden_size,
                                           bias=False, dtype=dtype,
                                           init_std=init_std)
            self.down = RowParallelLinear(ffn_hidden_size, hidden_size,
                                          bias=True, dtype=dtype,
                                          init_std=output_layer_init_std)
        # bias folded into the fused bias_gelu epilogue
        mp = get_hcg().get_model_parallel_world_size()
        self.up_bias = nn.Parameter(
            torch.zeros(ffn_hidden_size // mp, dtype=dtype))
        self.up_bias.is_mp = True
        self.up_bias.partition_dim = 0

    def forward(self, x):
        return self.down(bias_gelu(self.up(x), self.up_bias))


class TransformerDecoderLayer(nn.Module):
    """Pre-LN decoder layer (hybrid_model.py:489-674); MoE branch optional."""

    def __init__(self, hidden_size: int, num_heads: int, ffn_hidden_size: int,
                 hidden_dropout: float = 0.0, attn_dropout: float = 0.0,
                 fused_attn: bool = True, dtype: Optional[torch.dtype] = None,
                 sequence_parallel: bool = False, init_std: float = 0.02,
                 num_layers_for_scale: int = 1, expert_module: Optional[nn.Module] = None,
                 recompute_granularity: str = "full", use_recompute: bool = False,
                 cp_backend: str = "ulysses", cp_zigzag: bool = False):
        super().__init__()
        out_std = init_std / math.sqrt(2.0 * num_layers_for_scale)
        self.ln1 = FusedLayerNorm(hidden_size, dtype=dtype)
        self.ln2 = FusedLayerNorm(hidden_size, dtype=dtype)
        self.sequence_parallel = sequence_parallel
        self.hidden_dropout_p = hidden_dropout
        self.use_recompute = use_recompute
        self.recompute_granularity = recompute_granularity
        if sequence_parallel:
            sp_ops.mark_as_sp_param(self.ln1.weight)
            sp_ops.mark_as_sp_param(self.ln1.bias)
            sp_ops.mark_as_sp_param(self.ln2.weight)
            sp_ops.mark_as_sp_param(self.ln2.bias)
        self.attn = MultiHeadAttention(hidden_size, num_heads,
                                       attn_dropout=attn_dropout,
                                       fused_attn=fused_attn, dtype=dtype,
                                       sequence_parallel=sequence_parallel,
                                       init_std=init_std,
                                       output_layer_init_std=out_std,
                                       cp_backend=cp_backend,
                                       cp_zigzag=cp_zigzag)
        if expert_module is not None:
            self.ffn = expert_module
        else:
            self.ffn = FFN(hidden_size, ffn_hidden_size, dtype=dtype,
                           init_std=init_std, output_layer_init_std=out_std,
                           sequence_parallel=sequence_parallel)

    def _dropout(self, x):
        if self.hidden_dropout_p > 0.0 and self.training:
            return F.dropout(x, p=self.hidden_dropout_p)
        return x

    def _attn_branch(self, x, cache=None, use_cache=False):
        """dropout(attn(ln1(x))) — the branch WITHOUT the residual add
        (the add is fused into ln2, see forward)."""
        h = self.ln1(x)
        if self.use_recompute and self.recompute_granularity == "core_attn" \
                and self.training and cache is None and not use_cache \
                and torch.is_grad_enabled():
            # recompute only the attention core (QK^T/softmax/PV + projs);
            # reference granularity "core_attn" (hybrid_model.py:303-346)
            a = checkpoint(lambda t: self.attn(t)[0], h,
                           use_reentrant=False,
                           context_fn=checkpoint_rng_context)
            return self._dropout(a), None
        a, new_cache = self.attn(h, cache=cache, use_cache=use_cache)
        return self._dropout(a), new_cache

    def _attn_branch_nocache(self, x):
        return self._attn_branch(x)[0]

    def forward(self, x, cache: Optional[KVCache] = None, use_cache: bool = False):
        if self.use_recompute and self.recompute_granularity == "full_attn" \
                and self.training and not use_cache and torch.is_grad_enabled():
            a = checkpoint(self._attn_branch_nocache, x,
                           use_reentrant=False,
                           context_fn=checkpoint_rng_context)
            new_cache = None
        else:
            a, new_cache = self._attn_branch(x, cache, use_cache)
        # fused: h2 = LN(x + a), y = x + a (one kernel; the backward joins
        # the residual gradient in the same pass)
        from paddlefleetx_amd.ops import layernorm_residual
        h2, y = layernorm_residual(a, x, self.ln2.weight, self.ln2.bias,
                                   self.ln2.eps)
        z = y + self._dropout(self.ffn(h2))
        if use_cache:
            return z, new_cache
        return z


class GPTEmbeddings(nn.Module):
    """Vocab-parallel word embedding + learned positions (hybrid_model.py:682-736)."""

    def __init__(self, vocab_size: int, hidden_size: int,
                 max_position_embeddings: int, dropout: float = 0.0,
                 dtype: Optional[torch.dtype] = None,
                 sequence_parallel: bool = False, init_std: float = 0.02):
        super().__init__()
        self.word_embeddings = VocabParallelEmbedding(vocab_size, hidden_size,
                                                      dtype=dtype,
                                                      init_std=init_std)
        self.position_embeddings = nn.Embedding(max_position_embeddings,
                                                hidden_size, dtype=dtype)
        # REPLICATED across mp ranks -> must init from the global stream,
        # not the per-mp-rank local_seed (else SP's seq allgather mixes
        # inconsistent copies; reference env.py:34-98 seed discipline)
        nn.init.normal_(self.position_embeddings.weight, std=init_std)
        self.dropout_p = dropout
        self.sequence_parallel = sequence_parallel

    def forward(self, input_ids, position_ids=None):
        B, S = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(S, device=input_ids.device).unsqueeze(0)
        x = self.word_embeddings(input_ids) + self.position_embeddings(position_ids)
        if self.dropout_p > 0.0 and self.training:
            x = F.dropout(x, p=self.dropout_p)
        if self.sequence_parallel:
            x = sp_ops.scatter_to_sp_region(x)
        return x


class GPTModel(nn.Module):
    """Decoder stack. Activations [B, S, H] ([s/mp, B, H] under SP)."""

    def __init__(self, vocab_size: int = 50304, hidden_size: int = 1024,
                 num_layers: int = 24, num_attention_heads: int = 16,
                 ffn_hidden_size: Optional[int] = None,
                 max_position_embeddings: int = 1024,
                 hidden_dropout_prob: float = 0.0,
                 attention_probs_dropout_prob: float = 0.0,
                 fused_attn: bool = True, use_recompute: bool = False,
                 recompute_granularity: str = "full",
                 sequence_parallel: bool = False,
                 initializer_range: float = 0.02,
                 moe_configs: Optional[Dict[str, Any]] = None,
                 cp_backend: str = "ulysses", cp_zigzag: bool = False,
                 dtype: Optional[torch.dtype] = None, **unused: Any):
        super().__init__()
        ffn_hidden_size = ffn_hidden_size or 4 * hidden_size

        def _make_expert_module():
            """MoE FFN when expert_mode (hybrid_model.py:535-587)."""
            if not (moe_configs and moe_configs.get("expert_mode", False)):
                return None
            from paddlefleetx_amd.models.moe import MoELayer
            return MoELayer(
                hidden_size, ffn_hidden_size,
                num_experts=moe_configs.get("num_experts", 1),
                gate=moe_configs.get("gate", "gshard"),
                top_k=moe_configs.get("top_k", 2),
                capacity_factor=moe_configs.get("capacity_factor"),
                dtype=dtype)
        self.use_recompute = use_recompute
        self.recompute_granularity = recompute_granularity
        self.sequence_parallel = sequence_parallel
        self.embeddings = GPTEmbeddings(vocab_size, hidden_size,
                                        max_position_embeddings,
                                        dropout=hidden_dropout_prob, dtype=dtype,
                                        sequence_parallel=sequence_parallel,
                                        init_std=initializer_range)
        self.layers = nn.ModuleList([
            TransformerDecoderLayer(hidden_size, num_attention_heads,
                                    ffn_hidden_size,
                                    hidden_dropout=hidden_dropout_prob,
                                    attn_dropout=attention_probs_dropout_prob,
                                    fused_attn=fused_attn, dtype=dtype,
                                    sequence_parallel=sequence_parallel,
                                    init_std=initializer_range,
                                    num_layers_for_scale=num_layers,
                                    use_recompute=use_recompute,
                                    recompute_granularity=recompute_granularity,
                                    expert_module=_make_expert_module(),
                                    cp_backend=cp_backend,
                                    cp_zigzag=cp_zigzag)
            for _ in range(num_layers)])
        self.final_ln = FusedLayerNorm(hidden_size, dtype=dtype)
        if sequence_parallel:
            sp_ops.mark_as_sp_param(self.final_ln.weight)
            sp_ops.mark_as_sp_param(self.final_ln.bias)

    def forward(self, input_ids, position_ids=None,
                caches: Optional[List[KVCache]] = None, use_cache: bool = False):
        x = self.embeddings(input_ids, position_ids)
        new_caches: List[KVCache] = []
        for i, layer in enumerate(self.layers):
            cache_i = caches[i] if caches is not None else None
            if (self.use_recompute and self.recompute_granularity == "full"
                    and self.training and not use_cache
                    and torch.is_grad_enabled()):
                x = checkpoint(layer, x, use_reentrant=False,
                               context_fn=checkpoint_rng_context)
            elif use_cache:
                x, c = layer(x, cache=cache_i, use_cache=True)
                new_caches.append(c)
            else:
                x = layer(x, cache=cache_i)
        x = self.final_ln(x)
        if self.sequence_parallel:
            x = sp_ops.gather_from_sp_region(x)
        if use_cache:
            return x, new_caches
        return x


class GPTForPretraining(nn.Module):
    """Adds tied-embedding parallel logits (hybrid_model.py:897-941)."""

    def __init__(self, gpt: GPTModel):
        super().__init__()
        self.gpt = gpt

    def forward(self, input_ids, position_ids=None, caches=None,
                use_cache: bool = False):
        out = self.gpt(input_ids, position_ids, caches=caches,
                       use_cache=use_cache)
        if use_cache:
            hidden, new_caches = out
        else:
            hidden = out
        logits = parallel_matmul(hidden, self.gpt.embeddings.word_embeddings.weight,
                                 parallel_output=True)
        if use_cache:
            return logits, new_caches
        return logits


class GPTForSequenceClassification(nn.Module):
    """GPT + classification head on the last real token
    (single_model.py:856-897)."""

    def __init__(self, gpt: GPTModel, num_classes: int = 2):
        super().__init__()
        self.gpt = gpt
        hidden = gpt.final_ln.weight.shape[0]
        self.score = nn.Linear(hidden, num_classes, bias=False,
                               dtype=gpt.final_ln.weight.dtype)

    def forward(self, input_ids, attention_mask=None, position_ids=None):
        hidden = self.gpt(input_ids, position_ids)
        logits = self.score(hidden)  # [B, S, C]
        if attention_mask is not None:
            last = attention_mask.long().sum(dim=-1).clamp(min=1) - 1
        else:
            last = torch.full((input_ids.shape[0],), input_ids.shape[1] - 1,
                              device=input_ids.device, dtype=torch.long)
        return logits[torch.arange(logits.shape[0], device=logits.device),
                      last]


class GPTPretrainingCriterion(nn.Module):
    """(Parallel) softmax CE with loss mask (hybrid_model.py:943-996).
    Under context parallel each rank scores its sequence shard and the
    masked sum/count are all-reduced so every rank sees the global loss."""

    def __init__(self, sequence_parallel: bool = False):
        super().__init__()
        self.ce = ParallelCrossEntropy()

    def forward(self, logits, labels, loss_mask=None):
        losses = self.ce(logits, labels)  # [N] fp32
        cp = get_hcg().get_context_parallel_world_size()
        if loss_mask is not None:
            m = loss_mask.reshape(-1).float()
            num, den = (losses * m).sum(), m.sum()
        else:
            num = losses.sum()
            den = torch.tensor(float(losses.numel()), device=losses.device)
        if cp > 1:
            from paddlefleetx_amd.parallel.cp import cp_allreduce_sum
            num = cp_allreduce_sum(num)
            den = cp_allreduce_sum(den)
        return num / den.clamp(min=1.0)
