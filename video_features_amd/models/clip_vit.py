"""Native CLIP image encoder (ViT-B/32 and ViT-B/16).

The reference imports OpenAI's ``clip`` pip package (reference
models/CLIP/extract_clip.py:46-63); this is a from-scratch implementation of
the same architecture: patch-embed conv → [CLS] + learned positional
embeddings → pre-LN transformer (MHSA + QuickGELU MLP) → final LN → linear
projection to the 512-d joint space.

Hot ops (LayerNorm, QuickGELU, the attention core) dispatch through
``video_features_amd.ops`` — hand-written CDNA4 HIP kernels on MI355X,
PyTorch reference implementations on CPU.  Plain GEMMs (qkv/proj/MLP) go to
rocBLAS/hipBLASLt via ``torch.nn.Linear``.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
from torch import nn

from .. import ops


@dataclass
class ViTConfig:
    input_resolution: int = 224
    patch_size: int = 32
    width: int = 768
    layers: int = 12
    heads: int = 12
    output_dim: int = 512


VIT_B32 = ViTConfig(patch_size=32)
VIT_B16 = ViTConfig(patch_size=16)


class LayerNorm(nn.LayerNorm):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.layer_norm(x, self.weight, self.bias, self.eps)


class MultiheadSelfAttention(nn.Module):
    def __init__(self, width: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = width // heads
        self.qkv = nn.Linear(width, 3 * width)
        self.proj = nn.Linear(width, width)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        o = ops.mhsa_fused(self.qkv(x), self.heads)
        return self.proj(o)


class MLP(nn.Module):
    def __init__(self, width: int):
        super().__init__()
        self.c_fc = nn.Linear(width, 4 * width)
        self.c_proj = nn.Linear(4 * width, width)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.c_proj(ops.quick_gelu(self.c_fc(x)))


class ResidualAttentionBlock(nn.Module):
    def __init__(self, width: int, heads: int):
        super().__init__()
        self.ln_1 = LayerNorm(width)
        self.attn = MultiheadSelfAttention(width, heads)
        self.ln_2 = LayerNorm(width)
        self.mlp = MLP(width)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class VisionTransformer(nn.Module):
    def __init__(self, cfg: ViTConfig = VIT_B32):
        super().__init__()
        self.cfg = cfg
        w = cfg.width
        self.conv1 = nn.Conv2d(3, w, kernel_size=cfg.patch_size,
                               stride=cfg.patch_size, bias=False)
        n_patches = (cfg.input_resolution // cfg.patch_size) ** 2
        scale = w ** -0.5
        self.class_embedding = nn.Parameter(scale * torch.randn(w))
        self.positional_embedding = nn.Parameter(
            scale * torch.randn(n_patches + 1, w))
        self.ln_pre = LayerNorm(w)
        self.blocks = nn.ModuleList(
            [ResidualAttentionBlock(w, cfg.heads) for _ in range(cfg.layers)])
        self.ln_post = LayerNorm(w)
        self.proj = nn.Parameter(scale * torch.randn(w, cfg.output_dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """(T, 3, R, R) preprocessed frames → (T, output_dim) features."""
        x = self.conv1(x)                               # (T, W, R/p, R/p)
        x = x.flatten(2).transpose(1, 2)                # (T, P, W)
        cls = self.class_embedding.to(x.dtype).expand(x.shape[0], 1, -1)
        x = torch.cat([cls, x], dim=1)
        x = x + self.positional_embedding.to(x.dtype)
        x = self.ln_pre(x)
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_post(x[:, 0, :])
        return x @ self.proj.to(x.dtype)

    # reference parity name (reference extract_clip.py:128 uses
    # model.encode_image)
    def encode_image(self, x: torch.Tensor) -> torch.Tensor:
        return self.forward(x)


def build_clip_vit(feature_type: str) -> VisionTransformer:
    if feature_type in ('CLIP-ViT-B/32', 'CLIP4CLIP-ViT-B-32'):
        return VisionTransformer(VIT_B32)
    if feature_type == 'CLIP-ViT-B/16':
        return VisionTransformer(VIT_B16)
    raise ValueError(f'unknown CLIP feature type {feature_type!r}')
