"""Preprocessing transforms for every model family.

One shared module replaces the reference's three copy-pasted transform files
(reference models/i3d/transforms/transforms.py, models/raft/transforms/
transforms.py, models/r21d/transforms/rgb_transforms.py).  All transforms are
tensor-native (no PIL round-trips) so the same code runs on CPU and fuses
into the GPU pipeline.

Conventions: image batches are float32/bf16 ``(T, C, H, W)``; video tensors
for 3D nets are ``(C, T, H, W)``.
"""
from __future__ import annotations

import math
from typing import Sequence

import torch
import torch.nn.functional as F


def resize_improved(frames: torch.Tensor, size: int,
                    smaller_edge: bool = True) -> torch.Tensor:
    """Aspect-preserving resize of ``(T, C, H, W)`` so the smaller (or larger)
    edge equals ``size`` (reference i3d transforms.py:87-137 semantics)."""
    h, w = frames.shape[-2:]
    edge = min(h, w) if smaller_edge else max(h, w)
    if edge == size:
        return frames
    scale = size / edge
    nh, nw = int(math.floor(h * scale)), int(math.floor(w * scale))
    # guarantee the constrained edge hits exactly `size`
    if smaller_edge:
        if h < w:
            nh = size
        else:
            nw = size
    else:
        if h > w:
            nh = size
        else:
            nw = size
    return F.interpolate(frames, size=(nh, nw), mode='bilinear',
                         align_corners=False)


def center_crop(frames: torch.Tensor, size: int) -> torch.Tensor:
    """Center crop the last two dims to (size, size)
    (reference i3d transforms.py:7-18)."""
    h, w = frames.shape[-2:]
    top = (h - size) // 2
    left = (w - size) // 2
    return frames[..., top:top + size, left:left + size]


def scale_to_pm1(frames: torch.Tensor) -> torch.Tensor:
    """uint8-range [0,255] → [-1, 1] (reference i3d transforms.py:21-24)."""
    return frames * (2.0 / 255.0) - 1.0


def clamp(frames: torch.Tensor, lo: float, hi: float) -> torch.Tensor:
    return frames.clamp(lo, hi)


def flow_to_uint8_range(flow: torch.Tensor, bound: float = 20.0) -> torch.Tensor:
    """Quantize flow in [-bound, bound] exactly as the reference does before
    feeding the I3D flow stream (reference i3d transforms.py:43-51):
    ``round(128 + 255/(2*bound) * x)`` clamped to [0, 255]."""
    x = torch.round(128.0 + (255.0 / (2.0 * bound)) * flow)
    return x.clamp(0.0, 255.0)


def normalize(frames: torch.Tensor, mean: Sequence[float],
              std: Sequence[float]) -> torch.Tensor:
    """Channel-wise normalization on (..., C, H, W)."""
    mean_t = torch.as_tensor(mean, dtype=frames.dtype, device=frames.device)
    std_t = torch.as_tensor(std, dtype=frames.dtype, device=frames.device)
    return (frames - mean_t[:, None, None]) / std_t[:, None, None]


# ---- standard stats ------------------------------------------------------
IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)
CLIP_MEAN = (0.48145466, 0.4578275, 0.40821073)
CLIP_STD = (0.26862954, 0.26130258, 0.27577711)
KINETICS_MEAN = (0.43216, 0.394666, 0.37645)
KINETICS_STD = (0.22803, 0.22145, 0.216989)


def frames_uint8_to_float(frames_u8: torch.Tensor) -> torch.Tensor:
    """(T, H, W, 3) uint8 → (T, 3, H, W) float32 in [0, 1]."""
    return frames_u8.permute(0, 3, 1, 2).float() / 255.0


def clip_preprocess(frames_u8: torch.Tensor, size: int = 224) -> torch.Tensor:
    """CLIP's published preprocessing: bicubic resize of the smaller edge to
    ``size``, center crop, CLIP normalization.  Input (T, H, W, 3) uint8."""
    x = frames_u8.permute(0, 3, 1, 2).float() / 255.0
    h, w = x.shape[-2:]
    scale = size / min(h, w)
    nh, nw = max(size, int(round(h * scale))), max(size, int(round(w * scale)))
    x = F.interpolate(x, size=(nh, nw), mode='bicubic', align_corners=False)
    x = center_crop(x, size)
    return normalize(x, CLIP_MEAN, CLIP_STD)


def imagenet_preprocess(frames_u8: torch.Tensor, size: int = 224) -> torch.Tensor:
    """torchvision-style eval preprocessing used by the ResNet extractor
    (reference extract_resnet.py:38-44): resize smaller edge to 256,
    center-crop 224, ImageNet normalization."""
    x = frames_u8.permute(0, 3, 1, 2).float() / 255.0
    x = resize_improved(x, 256, smaller_edge=True)
    x = center_crop(x, size)
    return normalize(x, IMAGENET_MEAN, IMAGENET_STD)


def r21d_preprocess(frames_u8: torch.Tensor) -> torch.Tensor:
    """R(2+1)D video preprocessing (reference r21d rgb_transforms.py usage at
    extract_r21d.py:36-41): /255, resize to (128, 171), Kinetics norm,
    center crop 112.  Input (T, H, W, 3) uint8 → (3, T, 112, 112)."""
    x = frames_u8.permute(0, 3, 1, 2).float() / 255.0
    x = F.interpolate(x, size=(128, 171), mode='bilinear', align_corners=False)
    x = normalize(x, KINETICS_MEAN, KINETICS_STD)
    x = center_crop(x, 112)
    return x.permute(1, 0, 2, 3)   # (C, T, H, W)


def i3d_rgb_preprocess(frames_u8: torch.Tensor, resize_to: int = 256,
                       crop: int = 224) -> torch.Tensor:
    """I3D RGB stream preprocessing (reference extract_i3d.py:55-63):
    smaller-edge resize to 256, center crop 224, scale to [-1, 1].
    Input (T, H, W, 3) uint8 → (T, 3, crop, crop)."""
    x = frames_u8.permute(0, 3, 1, 2).float()
    x = resize_improved(x, resize_to, smaller_edge=True)
    x = center_crop(x, crop)
    return scale_to_pm1(x)


def i3d_flow_preprocess(flow: torch.Tensor, crop: int = 224,
                        bound: float = 20.0) -> torch.Tensor:
    """I3D flow stream preprocessing (reference extract_i3d.py:66-74):
    clamp ±20 → quantize to uint8 range → scale to [-1, 1].
    Input (T, 2, H, W) float flow."""
    x = clamp(flow, -bound, bound)
    x = flow_to_uint8_range(x, bound)
    x = center_crop(x, crop)
    return scale_to_pm1(x)
