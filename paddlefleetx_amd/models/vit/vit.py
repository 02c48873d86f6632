"""Vision Transformer, MI355X-native.

Reference: ppfleetx/models/vision_model/vit/vit.py (ViT :166, FusedBlock
:54-114 using paddle's FusedMultiHeadAttention/FusedFeedForward, factories
:422-556) plus layers/{attention,embedding,mlp,droppath}.py.

Here a single Block implementation uses the same fused kernels as the GPT
path: FusedLayerNorm (gfx950 HIP), hipBLASLt GEMMs, fused bias-gelu, and
bidirectional SDPA. The reference's fused-vs-plain checkpoint duality
(vit.py:301-420 state-dict converters) disappears: there is one layout.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.ops import FusedLayerNorm, bias_gelu

__all__ = [
    "ViT", "build_vit", "DropPath", "ViTAttention", "ViTMLP", "ViTPatchEmbed",
    "ViT_tiny_patch16_224", "ViT_base_patch16_224", "ViT_base_patch16_384",
    "ViT_base_patch32_224", "ViT_base_patch32_384", "ViT_large_patch16_224",
    "ViT_large_patch16_384", "ViT_large_patch32_224", "ViT_huge_patch14_224",
    "ViT_huge_patch14_384", "ViT_g_patch14_224", "ViT_G_patch14_224",
    "ViT_6B_patch14_224",
]


class DropPath(nn.Module):
    """Stochastic depth (reference layers/droppath.py)."""

    def __init__(self, drop_prob: float = 0.0):
        super().__init__()
        self.drop_prob = drop_prob

    def forward(self, x):
        if self.drop_prob == 0.0 or not self.training:
            return x
        keep = 1.0 - self.drop_prob
        mask = torch.rand(x.shape[0], *([1] * (x.ndim - 1)),
                          device=x.device, dtype=x.dtype) < keep
        return x * mask / keep


class ViTPatchEmbed(nn.Module):
    """Conv patchify (reference layers/embedding.py ViTPatchEmbed)."""

    def __init__(self, img_size: int = 224, patch_size: int = 16,
                 in_chans: int = 3, embed_dim: int = 768,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        assert img_size % patch_size == 0
        self.num_patches = (img_size // patch_size) ** 2
        self.proj = nn.Conv2d(in_chans, embed_dim, kernel_size=patch_size,
                              stride=patch_size, dtype=dtype)

    def forward(self, x):
        x = self.proj(x)                       # [B, E, H/p, W/p]
        return x.flatten(2).transpose(1, 2)    # [B, N, E]


class ViTAttention(nn.Module):
    """Pre-LN-free MHA (norm lives in the Block); bidirectional."""

    def __init__(self, dim: int, num_heads: int, qkv_bias: bool = True,
                 qk_scale: Optional[float] = None, attn_drop: float = 0.0,
                 proj_drop: float = 0.0, dtype: Optional[torch.dtype] = None):
        super().__init__()
        assert dim % num_heads == 0
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.scale = qk_scale or 1.0 / math.sqrt(self.head_dim)
        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias, dtype=dtype)
        self.proj = nn.Linear(dim, dim, dtype=dtype)
        self.attn_drop_p = attn_drop
        self.proj_drop_p = proj_drop

    def forward(self, x):
        B, N, C = x.shape
        qkv = self.qkv(x).view(B, N, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.unbind(dim=2)            # [B, N, h, D]
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        use_kernel = (x.is_cuda and x.dtype == torch.bfloat16
                      and self.head_dim in (64, 128)
                      and not (self.attn_drop_p > 0.0 and self.training))
        if use_kernel:
            from paddlefleetx_amd.ops import flash_attention
            o = flash_attention(q, k, v, causal=False, scale=self.scale)
            o = o.transpose(1, 2).reshape(B, N, C)
        else:
            scores = torch.matmul(q, k.transpose(-1, -2)) * self.scale
            probs = F.softmax(scores.float(), dim=-1).to(x.dtype)
            if self.attn_drop_p > 0.0 and self.training:
                probs = F.dropout(probs, p=self.attn_drop_p)
            o = torch.matmul(probs, v).transpose(1, 2).reshape(B, N, C)
        o = self.proj(o)
        if self.proj_drop_p > 0.0 and self.training:
            o = F.dropout(o, p=self.proj_drop_p)
        return o


class ViTMLP(nn.Module):
    """fc1 -> fused bias-gelu -> fc2 (reference layers/mlp.py ViTMLP)."""

    def __init__(self, dim: int, hidden: int, drop: float = 0.0,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.fc1 = nn.Linear(dim, hidden, bias=False, dtype=dtype)
        self.fc1_bias = nn.Parameter(torch.zeros(hidden, dtype=dtype))
        self.fc2 = nn.Linear(hidden, dim, dtype=dtype)
        self.drop_p = drop

    def forward(self, x):
        x = bias_gelu(self.fc1(x), self.fc1_bias)
        if self.drop_p > 0.0 and self.training:
            x = F.dropout(x, p=self.drop_p)
        x = self.fc2(x)
        if self.drop_p > 0.0 and self.training:
            x = F.dropout(x, p=self.drop_p)
        return x

    @property
    def bias_view(self):
        return self.fc1_bias


class Block(nn.Module):
    """Pre-LN transformer encoder block (vit.py:116-163)."""

    def __init__(self, dim, num_heads, mlp_ratio=4.0, qkv_bias=False,
                 qk_scale=None, drop=0.0, attn_drop=0.0, drop_path=0.0,
                 epsilon=1e-5, dtype=None):
        super().__init__()
        self.norm1 = FusedLayerNorm(dim, eps=epsilon, dtype=dtype)
        self.attn = ViTAttention(dim, num_heads, qkv_bias=qkv_bias,
                                 qk_scale=qk_scale, attn_drop=attn_drop,
                                 proj_drop=drop, dtype=dtype)
        self.drop_path = DropPath(drop_path)
        self.norm2 = FusedLayerNorm(dim, eps=epsilon, dtype=dtype)
        self.mlp = ViTMLP(dim, int(dim * mlp_ratio), drop=drop, dtype=dtype)

    def forward(self, x):
        x = x + self.drop_path(self.attn(self.norm1(x)))
        x = x + self.drop_path(self.mlp(self.norm2(x)))
        return x


class ViT(nn.Module):
    """Vision Transformer (vit.py:166-420)."""

    def __init__(self, img_size=224, patch_size=16, in_chans=3,
                 class_num=1000, embed_dim=768, depth=12, num_heads=12,
                 mlp_ratio=4, qkv_bias=False, qk_scale=None, drop_rate=0.0,
                 attn_drop_rate=0.0, drop_path_rate=0.0, epsilon=1e-5,
                 representation_size=None, dtype=None, **unused):
        super().__init__()
        self.class_num = class_num
        self.representation_size = representation_size
        self.num_features = self.embed_dim = embed_dim

        self.patch_embed = ViTPatchEmbed(img_size, patch_size, in_chans,
                                         embed_dim, dtype=dtype)
        num_patches = self.patch_embed.num_patches
        self.pos_embed = nn.Parameter(
            torch.empty(1, num_patches + 1, embed_dim, dtype=dtype))
        self.cls_token = nn.Parameter(
            torch.zeros(1, 1, embed_dim, dtype=dtype))
        self.pos_drop_p = drop_rate

        dpr = torch.linspace(0, drop_path_rate, depth).tolist()
        self.blocks = nn.ModuleList([
            Block(embed_dim, num_heads, mlp_ratio=mlp_ratio,
                  qkv_bias=qkv_bias, qk_scale=qk_scale, drop=drop_rate,
                  attn_drop=attn_drop_rate, drop_path=dpr[i],
                  epsilon=epsilon, dtype=dtype)
            for i in range(depth)])
        self.norm = FusedLayerNorm(embed_dim, eps=epsilon, dtype=dtype)

        if representation_size is not None:
            self.head0 = nn.Linear(embed_dim, representation_size, dtype=dtype)
            self.tanh = nn.Tanh()
            self.head = nn.Linear(representation_size, class_num, dtype=dtype) \
                if class_num > 0 else nn.Identity()
            nn.init.xavier_uniform_(self.head0.weight)
            nn.init.zeros_(self.head0.bias)
            nn.init.xavier_uniform_(self.head.weight)
            nn.init.constant_(self.head.bias, -10.0)  # minus_tens_ (vit.py:240)
        else:
            self.head = nn.Linear(embed_dim, class_num, dtype=dtype) \
                if class_num > 0 else nn.Identity()
            nn.init.zeros_(self.head.weight)
            nn.init.zeros_(self.head.bias)

        nn.init.normal_(self.pos_embed, std=0.02)
        for m in self.modules():
            if isinstance(m, ViTAttention):
                nn.init.xavier_uniform_(m.qkv.weight)
                nn.init.xavier_uniform_(m.proj.weight)
                if m.qkv.bias is not None:
                    nn.init.zeros_(m.qkv.bias)
                nn.init.zeros_(m.proj.bias)
            elif isinstance(m, ViTMLP):
                nn.init.xavier_uniform_(m.fc1.weight)
                nn.init.xavier_uniform_(m.fc2.weight)
                nn.init.normal_(m.fc1_bias, std=1e-6)
                nn.init.normal_(m.fc2.bias, std=1e-6)

    def forward_features(self, x):
        B = x.shape[0]
        x = self.patch_embed(x)
        cls = self.cls_token.expand(B, -1, -1)
        x = torch.cat((cls, x), dim=1) + self.pos_embed
        if self.pos_drop_p > 0.0 and self.training:
            x = F.dropout(x, p=self.pos_drop_p)
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x)
        return x[:, 0]

    def forward(self, x):
        x = self.forward_features(x)
        if self.representation_size is not None:
            x = self.tanh(self.head0(x))
        return self.head(x)


def _factory(**base):
    def f(**kw):
        cfg = dict(base)
        cfg.update(kw)
        return ViT(**cfg)
    return f


# reference vit.py:422-556 factory sizes
ViT_tiny_patch16_224 = _factory(patch_size=16, embed_dim=192, depth=12,
                                num_heads=3, mlp_ratio=4, qkv_bias=True,
                                epsilon=1e-6, representation_size=192)
ViT_base_patch16_224 = _factory(patch_size=16, embed_dim=768, depth=12,
                                num_heads=12, mlp_ratio=4, qkv_bias=True,
                                epsilon=1e-6, representation_size=768)
ViT_base_patch16_384 = _factory(img_size=384, patch_size=16, embed_dim=768,
                                depth=12, num_heads=12, mlp_ratio=4,
                                qkv_bias=True, epsilon=1e-6)
ViT_base_patch32_224 = _factory(patch_size=32, embed_dim=768, depth=12,
                                num_heads=12, mlp_ratio=4, qkv_bias=True,
                                epsilon=1e-6, representation_size=768)
ViT_base_patch32_384 = _factory(img_size=384, patch_size=32, embed_dim=768,
                                depth=12, num_heads=12, mlp_ratio=4,
                                qkv_bias=True, epsilon=1e-6)
ViT_large_patch16_224 = _factory(patch_size=16, embed_dim=1024, depth=24,
                                 num_heads=16, mlp_ratio=4, qkv_bias=True,
                                 epsilon=1e-6, representation_size=1024)
ViT_large_patch16_384 = _factory(img_size=384, patch_size=16, embed_dim=1024,
                                 depth=24, num_heads=16, mlp_ratio=4,
                                 qkv_bias=True, epsilon=1e-6)
ViT_large_patch32_224 = _factory(patch_size=32, embed_dim=1024, depth=24,
                                 num_heads=16, mlp_ratio=4, qkv_bias=True,
                                 epsilon=1e-6, representation_size=1024)
ViT_huge_patch14_224 = _factory(patch_size=14, embed_dim=1280, depth=32,
                                num_heads=16, mlp_ratio=4, qkv_bias=True,
                                epsilon=1e-6, representation_size=1280)
ViT_huge_patch14_384 = _factory(img_size=384, patch_size=14, embed_dim=1280,
                                depth=32, num_heads=16, mlp_ratio=4,
                                qkv_bias=True, epsilon=1e-6)
ViT_g_patch14_224 = _factory(patch_size=14, embed_dim=1408, depth=40,
                             num_heads=16, mlp_ratio=4.364, qkv_bias=True,
                             epsilon=1e-6)
ViT_G_patch14_224 = _factory(patch_size=14, embed_dim=1664, depth=48,
                             num_heads=16, mlp_ratio=4.9231, qkv_bias=True,
                             epsilon=1e-6)
ViT_6B_patch14_224 = _factory(patch_size=14, embed_dim=2320, depth=80,
                              num_heads=16, mlp_ratio=4.955, qkv_bias=True,
                              epsilon=1e-6)

_FACTORIES = {
    "ViT_tiny_patch16_224": ViT_tiny_patch16_224,
    "ViT_base_patch16_224": ViT_base_patch16_224,
    "ViT_base_patch16_384": ViT_base_patch16_384,
    "ViT_base_patch32_224": ViT_base_patch32_224,
    "ViT_base_patch32_384": ViT_base_patch32_384,
    "ViT_large_patch16_224": ViT_large_patch16_224,
    "ViT_large_patch16_384": ViT_large_patch16_384,
    "ViT_large_patch32_224": ViT_large_patch32_224,
    "ViT_huge_patch14_224": ViT_huge_patch14_224,
    "ViT_huge_patch14_384": ViT_huge_patch14_384,
    "ViT_g_patch14_224": ViT_g_patch14_224,
    "ViT_G_patch14_224": ViT_G_patch14_224,
    "ViT_6B_patch14_224": ViT_6B_patch14_224,
}


def build_vit(name: str, **kw) -> ViT:
    if name == "ViT":
        return ViT(**kw)
    if name not in _FACTORIES:
        raise ValueError(f"unknown ViT variant {name!r}")
    return _FACTORIES[name](**kw)
