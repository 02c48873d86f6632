"""Template embedding: per-template pair stack + attention over templates.

Reference: ppfleetx/models/protein_folding/template.py (TemplatePair :36
— a pair-only Evoformer block; SingleTemplateEmbedding :164 — distogram
+ backbone-frame features -> pair channels; TemplateEmbedding :290 —
cross-attention of the query pair rep over the template embeddings).
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn

from paddlefleetx_amd.models.protein_folding.evoformer import (
    GatedAttention, Transition, TriangleAttention, TriangleMultiplication)
from paddlefleetx_amd.models.protein_folding.geometry import Rigid


class TemplatePair(nn.Module):
    """Pair-only trunk block (triangle ops + transition), applied per
    template (reference template.py:36-161)."""

    def __init__(self, pair_dim: int, num_heads: int = 4, head_dim: int = 16):
        super().__init__()
        self.tri_attn_start = TriangleAttention(pair_dim, num_heads,
                                                head_dim, True)
        self.tri_attn_end = TriangleAttention(pair_dim, num_heads,
                                              head_dim, False)
        self.tri_mul_out = TriangleMultiplication(pair_dim, pair_dim, True)
        self.tri_mul_in = TriangleMultiplication(pair_dim, pair_dim, False)
        self.trans = Transition(pair_dim)

    def forward(self, z):
        z = z + self.tri_attn_start(z)
        z = z + self.tri_attn_end(z)
        z = z + self.tri_mul_out(z)
        z = z + self.tri_mul_in(z)
        z = z + self.trans(z)
        return z


def dgram_from_positions(pos: torch.Tensor, num_bins: int = 39,
                         min_bin: float = 3.25, max_bin: float = 50.75
                         ) -> torch.Tensor:
    """One-hot distance histogram of pairwise CB distances
    (reference template.py:190-230 featurization)."""
    d2 = ((pos[..., :, None, :] - pos[..., None, :, :]) ** 2).sum(-1)
    bins = torch.linspace(min_bin, max_bin, num_bins, device=pos.device) ** 2
    og = (d2[..., None] > bins).to(pos.dtype)
    return og * torch.cat([1.0 - og[..., 1:],
                           torch.ones_like(og[..., :1])], dim=-1)


class SingleTemplateEmbedding(nn.Module):
    """Distogram + unit-vector backbone features -> pair_dim channels,
    then `num_blocks` TemplatePair iterations."""

    def __init__(self, pair_dim: int = 64, num_bins: int = 39,
                 num_blocks: int = 2):
        super().__init__()
        # features: dgram + mask2d + 3 unit-vector comps + 1 frame mask
        self.proj = nn.Linear(num_bins + 1 + 3 + 1, pair_dim)
        self.blocks = nn.ModuleList(
            [TemplatePair(pair_dim) for _ in range(num_blocks)])
        self.norm = nn.LayerNorm(pair_dim)
        self.num_bins = num_bins

    def forward(self, cb_pos: torch.Tensor, frames: Rigid,
                frame_mask: torch.Tensor, mask_2d: torch.Tensor):
        # cb_pos [B, N, 3]; frames over [B, N]; mask_2d [B, N, N]
        dg = dgram_from_positions(cb_pos, self.num_bins)
        # inter-residue CA direction in each residue's local frame
        rel = frames.invert()[..., :, None].apply(
            cb_pos[..., None, :, :])                    # [B, N, N, 3]
        unit = rel / torch.linalg.norm(rel, dim=-1,
                                       keepdim=True).clamp_min(1e-6)
        fm2d = (frame_mask[..., :, None] * frame_mask[..., None, :])
        unit = unit * fm2d[..., None]
        feats = torch.cat([dg, mask_2d[..., None], unit, fm2d[..., None]],
                          dim=-1)
        z = self.proj(feats)
        for blk in self.blocks:
            z = blk(z)
        return self.norm(z)


class TemplateEmbedding(nn.Module):
    """Cross-attention of the query pair rep over per-template
    embeddings (reference template.py:290-368)."""

    def __init__(self, pair_dim: int = 64, template_dim: int = 64,
                 num_heads: int = 4, head_dim: int = 16):
        super().__init__()
        self.single = SingleTemplateEmbedding(template_dim)
        self.attn = GatedAttention(pair_dim, template_dim, num_heads,
                                   head_dim, pair_dim, gating=False)

    def forward(self, query_pair: torch.Tensor, template_batch: Dict,
                mask_2d: torch.Tensor) -> torch.Tensor:
        """query_pair [B, N, N, Cz]; template_batch holds per-template
        cb_pos [B, T, N, 3], frames (Rigid over [B, T, N]), frame_mask
        [B, T, N]."""
        cb = template_batch["cb_pos"]
        B, T, N = cb.shape[:3]
        embs = []
        for t in range(T):
            embs.append(self.single(cb[:, t], template_batch["frames"][:, t],
                                    template_batch["frame_mask"][:, t],
                                    mask_2d))
        emb = torch.stack(embs, dim=1)                  # [B, T, N, N, Ct]
        q = query_pair.view(B, N * N, 1, -1)            # query per (i,j)
        kv = emb.permute(0, 2, 3, 1, 4).reshape(B, N * N, T, -1)
        out = self.attn(q, kv)                          # [B, N*N, 1, Cz]
        return out.view(B, N, N, -1)
