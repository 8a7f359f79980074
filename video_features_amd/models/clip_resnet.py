"""CLIP's ModifiedResNet image encoder (RN50/RN101).

The reference's ``ExtractCLIP`` codes ResNet CLIP backbones without
exposing them in the CLI (reference models/CLIP/extract_clip.py:46-63 via
the ``clip`` package).  Here they are first-class feature
types (``CLIP-RN50``, ``CLIP-RN101``).

Architecture (OpenAI CLIP): a 3-conv anti-aliased stem (avg-pool instead
of max-pool / strided conv), Bottlenecks whose stride-2 is an avg-pool
before conv3 (and in the downsample branch), and a final
``AttentionPool2d`` — one multi-head attention step with a mean-pooled
query token and learned positional embeddings — producing the joint-space
embedding (1024-d for RN50).  Module names match the published state
dicts (``visual.`` stripped by utils.convert_checkpoints).
"""
from __future__ import annotations

from collections import OrderedDict

import torch
import torch.nn.functional as F
from torch import nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.relu1 = nn.ReLU(inplace=True)
        self.conv2 = nn.Conv2d(planes, planes, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.relu2 = nn.ReLU(inplace=True)
        self.avgpool = nn.AvgPool2d(stride) if stride > 1 else nn.Identity()
        self.conv3 = nn.Conv2d(planes, planes * self.expansion, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion)
        self.relu3 = nn.ReLU(inplace=True)
        self.downsample = None
        self.stride = stride
        if stride > 1 or inplanes != planes * self.expansion:
            # OpenAI ordering (and state-dict keys "-1"/"0"/"1"):
            # avgpool, 1x1 conv, bn
            self.downsample = nn.Sequential(OrderedDict([
                ('-1', nn.AvgPool2d(stride)),
                ('0', nn.Conv2d(inplanes, planes * self.expansion, 1,
                                stride=1, bias=False)),
                ('1', nn.BatchNorm2d(planes * self.expansion))]))

    def forward(self, x):
        identity = x
        out = self.relu1(self.bn1(self.conv1(x)))
        out = self.relu2(self.bn2(self.conv2(out)))
        out = self.avgpool(out)
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu3(out + identity)


class AttentionPool2d(nn.Module):
    def __init__(self, spacial_dim: int, embed_dim: int, num_heads: int,
                 output_dim: int = None):
        super().__init__()
        self.positional_embedding = nn.Parameter(
            torch.randn(spacial_dim ** 2 + 1, embed_dim) / embed_dim ** 0.5)
        self.k_proj = nn.Linear(embed_dim, embed_dim)
        self.q_proj = nn.Linear(embed_dim, embed_dim)
        self.v_proj = nn.Linear(embed_dim, embed_dim)
        self.c_proj = nn.Linear(embed_dim, output_dim or embed_dim)
        self.num_heads = num_heads

    def forward(self, x):
        b, c, h, w = x.shape
        x = x.flatten(2).permute(0, 2, 1)            # (B, HW, C)
        x = torch.cat([x.mean(dim=1, keepdim=True), x], dim=1)
        x = x + self.positional_embedding[None].to(x.dtype)
        q = self.q_proj(x[:, :1])                    # the pooled token only
        k = self.k_proj(x)
        v = self.v_proj(x)
        d = c // self.num_heads
        q = q.view(b, 1, self.num_heads, d).transpose(1, 2)
        k = k.view(b, -1, self.num_heads, d).transpose(1, 2)
        v = v.view(b, -1, self.num_heads, d).transpose(1, 2)
        attn = (q @ k.transpose(-2, -1)) / d ** 0.5
        o = attn.softmax(dim=-1) @ v                 # (B, H, 1, d)
        o = o.transpose(1, 2).reshape(b, 1, c)
        return self.c_proj(o)[:, 0]


class ModifiedResNet(nn.Module):
    def __init__(self, layers, output_dim: int, heads: int,
                 input_resolution: int = 224, width: int = 64):
        super().__init__()
        self.output_dim = output_dim
        self.input_resolution = input_resolution
        self.conv1 = nn.Conv2d(3, width // 2, 3, 2, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width // 2)
        self.relu1 = nn.ReLU(inplace=True)
        self.conv2 = nn.Conv2d(width // 2, width // 2, 3, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width // 2)
        self.relu2 = nn.ReLU(inplace=True)
        self.conv3 = nn.Conv2d(width // 2, width, 3, padding=1, bias=False)
        self.bn3 = nn.BatchNorm2d(width)
        self.relu3 = nn.ReLU(inplace=True)
        self.avgpool = nn.AvgPool2d(2)
        self._inplanes = width
        self.layer1 = self._make_layer(width, layers[0])
        self.layer2 = self._make_layer(width * 2, layers[1], stride=2)
        self.layer3 = self._make_layer(width * 4, layers[2], stride=2)
        self.layer4 = self._make_layer(width * 8, layers[3], stride=2)
        embed_dim = width * 32
        self.attnpool = AttentionPool2d(input_resolution // 32, embed_dim,
                                        heads, output_dim)

    def _make_layer(self, planes, blocks, stride=1):
        mods = [Bottleneck(self._inplanes, planes, stride)]
        self._inplanes = planes * Bottleneck.expansion
        for _ in range(1, blocks):
            mods.append(Bottleneck(self._inplanes, planes))
        return nn.Sequential(*mods)

    def forward(self, x):
        x = self.relu1(self.bn1(self.conv1(x)))
        x = self.relu2(self.bn2(self.conv2(x)))
        x = self.relu3(self.bn3(self.conv3(x)))
        x = self.avgpool(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.attnpool(x)

    # reference parity name
    def encode_image(self, x: torch.Tensor) -> torch.Tensor:
        return self.forward(x)


def build_clip_resnet(feature_type: str) -> ModifiedResNet:
    if feature_type == 'CLIP-RN50':
        return ModifiedResNet([3, 4, 6, 3], output_dim=1024, heads=32)
    if feature_type == 'CLIP-RN101':
        return ModifiedResNet([3, 4, 23, 3], output_dim=512, heads=32)
    raise ValueError(f'unknown CLIP ResNet variant {feature_type!r}')
