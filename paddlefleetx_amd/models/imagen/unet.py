"""Imagen U-Net (text-conditioned denoiser).

Reference: ppfleetx/models/multimodal_model/imagen/unet.py — SinusoidalPosEmb
:350, Block :382 (GroupNorm + scale-shift), ResnetBlock :400, CrossAttention
:464, TransformerBlock :723, Unet :858 and the sized variants in
modeling.py:36-93. Compact MI355X-native re-implementation: the attention
blocks run on hipBLASLt GEMMs; time/text conditioning follows the paper
(scale-shift GroupNorm + cross-attention on pooled + sequence text embeds).
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class SinusoidalPosEmb(nn.Module):
    def __init__(self, dim: int):
        super().__init__()
        self.dim = dim

    def forward(self, t):
        half = self.dim // 2
        freqs = torch.exp(-math.log(10000) *
                          torch.arange(half, device=t.device) / (half - 1))
        args = t[:, None].float() * freqs[None]
        return torch.cat([args.sin(), args.cos()], dim=-1)


class Block(nn.Module):
    """GroupNorm -> (scale, shift) -> SiLU -> conv (unet.py:382)."""

    def __init__(self, dim_in: int, dim_out: int, groups: int = 8):
        super().__init__()
        self.norm = nn.GroupNorm(groups, dim_in)
        self.conv = nn.Conv2d(dim_in, dim_out, 3, padding=1)

    def forward(self, x, scale_shift: Optional[Tuple] = None):
        x = self.norm(x)
        if scale_shift is not None:
            scale, shift = scale_shift
            x = x * (scale + 1) + shift
        return self.conv(F.silu(x))


class ResnetBlock(nn.Module):
    """Time+cond conditioned residual block (unet.py:400)."""

    def __init__(self, dim_in: int, dim_out: int, time_dim: int,
                 groups: int = 8):
        super().__init__()
        self.time_mlp = nn.Sequential(nn.SiLU(),
                                      nn.Linear(time_dim, dim_out * 2))
        self.block1 = Block(dim_in, dim_out, groups)
        self.block2 = Block(dim_out, dim_out, groups)
        self.res_conv = nn.Conv2d(dim_in, dim_out, 1) \
            if dim_in != dim_out else nn.Identity()

    def forward(self, x, t):
        # scale-shift conditions block2 (dim_out features), as in unet.py:400
        ss = self.time_mlp(t)[:, :, None, None].chunk(2, dim=1)
        h = self.block1(x)
        h = self.block2(h, scale_shift=ss)
        return h + self.res_conv(x)


class CrossAttention(nn.Module):
    """Image-queries x text-keys attention (unet.py:464)."""

    def __init__(self, dim: int, context_dim: int, heads: int = 8,
                 dim_head: int = 64):
        super().__init__()
        inner = heads * dim_head
        self.heads = heads
        self.scale = dim_head ** -0.5
        self.norm = nn.LayerNorm(dim)
        self.to_q = nn.Linear(dim, inner, bias=False)
        self.to_kv = nn.Linear(context_dim, inner * 2, bias=False)
        self.to_out = nn.Linear(inner, dim, bias=False)

    def forward(self, x, context, mask=None):
        # x: [B, N, C]; context: [B, T, Dc]
        B, N, _ = x.shape
        h = self.heads
        q = self.to_q(self.norm(x)).view(B, N, h, -1).transpose(1, 2)
        k, v = self.to_kv(context).chunk(2, dim=-1)
        k = k.view(B, -1, h, q.shape[-1]).transpose(1, 2)
        v = v.view(B, -1, h, q.shape[-1]).transpose(1, 2)
        sim = torch.matmul(q, k.transpose(-1, -2)) * self.scale
        if mask is not None:
            sim = sim.masked_fill(~mask[:, None, None, :].bool(), -1e4)
        attn = sim.float().softmax(dim=-1).to(x.dtype)
        out = torch.matmul(attn, v).transpose(1, 2).reshape(B, N, -1)
        return self.to_out(out)


class SelfAttention2d(nn.Module):
    def __init__(self, dim: int, heads: int = 8, dim_head: int = 64):
        super().__init__()
        inner = heads * dim_head
        self.heads = heads
        self.scale = dim_head ** -0.5
        self.norm = nn.GroupNorm(1, dim)
        self.to_qkv = nn.Conv2d(dim, inner * 3, 1, bias=False)
        self.to_out = nn.Conv2d(inner, dim, 1)

    def forward(self, x):
        B, C, H, W = x.shape
        qkv = self.to_qkv(self.norm(x)).view(B, 3, self.heads, -1, H * W)
        q, k, v = qkv.unbind(dim=1)          # [B, h, d, N]
        sim = torch.einsum("bhdn,bhdm->bhnm", q, k) * self.scale
        attn = sim.float().softmax(dim=-1).to(x.dtype)
        out = torch.einsum("bhnm,bhdm->bhdn", attn, v)
        return self.to_out(out.reshape(B, -1, H, W)) + x


class TransformerBlock(nn.Module):
    """Self-attn + cross-attn + FF at a U-Net resolution (unet.py:723)."""

    def __init__(self, dim: int, context_dim: int, heads: int = 8,
                 dim_head: int = 64, ff_mult: int = 2):
        super().__init__()
        self.self_attn = SelfAttention2d(dim, heads, dim_head)
        self.cross = CrossAttention(dim, context_dim, heads, dim_head)
        self.ff = nn.Sequential(nn.LayerNorm(dim),
                                nn.Linear(dim, dim * ff_mult),
                                nn.GELU(),
                                nn.Linear(dim * ff_mult, dim))

    def forward(self, x, context=None, mask=None):
        x = self.self_attn(x)
        if context is not None:
            B, C, H, W = x.shape
            seq = x.flatten(2).transpose(1, 2)          # [B, HW, C]
            seq = seq + self.cross(seq, context, mask)
            seq = seq + self.ff(seq)
            x = seq.transpose(1, 2).view(B, C, H, W)
        return x


def Downsample(dim_in, dim_out):
    return nn.Conv2d(dim_in, dim_out, 4, 2, 1)


def Upsample(dim_in, dim_out):
    return nn.Sequential(nn.Upsample(scale_factor=2, mode="nearest"),
                         nn.Conv2d(dim_in, dim_out, 3, padding=1))


class Unet(nn.Module):
    """Text-conditioned diffusion U-Net (unet.py:858).

    lowres_cond=True adds a low-resolution conditioning image channel-wise
    (the super-resolution units SRUnet256/1024)."""

    def __init__(self, dim: int = 128, dim_mults=(1, 2, 4, 8), channels=3,
                 cond_dim: Optional[int] = None, text_embed_dim: int = 512,
                 num_resnet_blocks=1, layer_attns=(False, False, True, True),
                 layer_cross_attns=(False, False, True, True),
                 attn_heads: int = 8, attn_dim_head: int = 64,
                 lowres_cond: bool = False, groups: int = 8,
                 memory_efficient: bool = False, **unused):
        super().__init__()
        self.channels = channels
        self.lowres_cond = lowres_cond
        in_ch = channels * (2 if lowres_cond else 1)
        dims = [dim] + [dim * m for m in dim_mults]
        time_dim = dim * 4
        cond_dim = cond_dim or dim

        self.init_conv = nn.Conv2d(in_ch, dim, 7, padding=3)
        self.time_mlp = nn.Sequential(
            SinusoidalPosEmb(dim), nn.Linear(dim, time_dim), nn.SiLU(),
            nn.Linear(time_dim, time_dim))
        # text conditioning: project sequence embeds + pooled into time_dim
        self.text_proj = nn.Linear(text_embed_dim, cond_dim)
        self.text_pool_mlp = nn.Sequential(
            nn.Linear(text_embed_dim, time_dim), nn.SiLU(),
            nn.Linear(time_dim, time_dim))

        if isinstance(num_resnet_blocks, int):
            num_resnet_blocks = [num_resnet_blocks] * len(dim_mults)
        if not isinstance(layer_attns, (tuple, list)):
            layer_attns = [bool(layer_attns)] * len(dim_mults)

        # Efficient U-Net (Imagen paper; reference unet.py memory_efficient
        # :898, 1139-1148, 1255): each stage DOWNSAMPLES FIRST so its
        # resnet/attention blocks run at the reduced resolution — the
        # activation-memory shape the SR stages need at 256/1024 px
        self.memory_efficient = memory_efficient
        self.downs = nn.ModuleList()
        self.ups = nn.ModuleList()
        n = len(dim_mults)
        for i in range(n):
            d_in, d_out = dims[i], dims[i + 1]
            if memory_efficient:
                pre = Downsample(d_in, d_out)
                blocks = nn.ModuleList([
                    ResnetBlock(d_out, d_out, time_dim, groups)
                    for b in range(num_resnet_blocks[i])])
                down = None
            else:
                pre = None
                blocks = nn.ModuleList([
                    ResnetBlock(d_in if b == 0 else d_out, d_out, time_dim,
                                groups)
                    for b in range(num_resnet_blocks[i])])
                down = Downsample(d_out, d_out) if i < n - 1 else None
            attn = TransformerBlock(d_out, cond_dim, attn_heads,
                                    attn_dim_head) if layer_attns[i] else None
            self.downs.append(nn.ModuleList([blocks, attn, down, pre]))

        mid = dims[-1]
        self.mid_block1 = ResnetBlock(mid, mid, time_dim, groups)
        self.mid_attn = TransformerBlock(mid, cond_dim, attn_heads,
                                         attn_dim_head)
        self.mid_block2 = ResnetBlock(mid, mid, time_dim, groups)

        for i in reversed(range(n)):
            d_in, d_out = dims[i + 1], dims[i]
            blocks = nn.ModuleList([
                ResnetBlock(d_in * 2 if b == 0 else d_in, d_in, time_dim,
                            groups)
                for b in range(num_resnet_blocks[i])])
            attn = TransformerBlock(d_in, cond_dim, attn_heads,
                                    attn_dim_head) if layer_attns[i] else None
            # memory_efficient pre-downsamples EVERY stage, so every up
            # stage upsamples (restoring full resolution before the head)
            up = Upsample(d_in, d_out) if (i > 0 or memory_efficient) \
                else None
            self.ups.append(nn.ModuleList([blocks, attn, up]))

        final_in = dims[0] if memory_efficient else dims[1]
        self.final_block = ResnetBlock(final_in, dim, time_dim, groups)
        self.final_conv = nn.Conv2d(dim, channels, 1)
        self._cross = list(layer_cross_attns)

    def forward(self, x, time, text_embeds=None, text_mask=None,
                lowres_cond_img=None):
        if self.lowres_cond:
            assert lowres_cond_img is not None
            x = torch.cat([x, lowres_cond_img], dim=1)
        x = self.init_conv(x)
        t = self.time_mlp(time)
        context = None
        if text_embeds is not None:
            context = self.text_proj(text_embeds)
            if text_mask is not None:
                pooled = (text_embeds * text_mask[..., None]).sum(1) / \
                    text_mask.sum(1, keepdim=True).clamp(min=1)
            else:
                pooled = text_embeds.mean(dim=1)
            t = t + self.text_pool_mlp(pooled)

        skips = []
        for blocks, attn, down, pre in self.downs:
            if pre is not None:
                x = pre(x)
            for b in blocks:
                x = b(x, t)
            if attn is not None:
                x = attn(x, context, text_mask)
            skips.append(x)
            if down is not None:
                x = down(x)

        x = self.mid_block1(x, t)
        x = self.mid_attn(x, context, text_mask)
        x = self.mid_block2(x, t)

        for blocks, attn, up in self.ups:
            skip = skips.pop()
            if x.shape[-2:] != skip.shape[-2:]:
                x = F.interpolate(x, size=skip.shape[-2:], mode="nearest")
            x = torch.cat([x, skip], dim=1)
            for b in blocks:
                x = b(x, t)
            if attn is not None:
                x = attn(x, context, text_mask)
            if up is not None:
                x = up(x)

        return self.final_conv(self.final_block(x, t))


# sized variants (modeling.py:36-93)
def Unet64_397M(**kw):
    cfg = dict(dim=192, dim_mults=(1, 2, 3, 4), num_resnet_blocks=3,
               layer_attns=(False, True, True, True),
               layer_cross_attns=(False, True, True, True))
    cfg.update(kw)
    return Unet(**cfg)


def BaseUnet64(**kw):
    cfg = dict(dim=512, dim_mults=(1, 2, 3, 4), num_resnet_blocks=3,
               layer_attns=(False, True, True, True),
               layer_cross_attns=(False, True, True, True))
    cfg.update(kw)
    return Unet(**cfg)


def SRUnet256(**kw):
    cfg = dict(dim=128, dim_mults=(1, 2, 4, 8), num_resnet_blocks=(2, 4, 8, 8),
               layer_attns=(False, False, False, True),
               layer_cross_attns=(False, False, False, True),
               lowres_cond=True, memory_efficient=True)
    cfg.update(kw)
    return Unet(**cfg)


def SRUnet1024(**kw):
    cfg = dict(dim=128, dim_mults=(1, 2, 4, 8), num_resnet_blocks=(2, 4, 8, 8),
               layer_attns=False, layer_cross_attns=(False, False, False, True),
               lowres_cond=True, memory_efficient=True)
    if not isinstance(cfg["layer_attns"], (tuple, list)):
        cfg["layer_attns"] = (False, False, False, False)
    cfg.update(kw)
    return Unet(**cfg)
