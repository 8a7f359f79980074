"""feature_type string → extractor class (lazy imports keep startup light)."""
from __future__ import annotations

from typing import Callable, Dict


def _clip():
    from ..extractors.clip import ExtractCLIP
    return ExtractCLIP


def _resnet():
    from ..extractors.resnet import ExtractResNet
    return ExtractResNet


def _r21d():
    from ..extractors.r21d import ExtractR21D
    return ExtractR21D


def _i3d():
    from ..extractors.i3d import ExtractI3D
    return ExtractI3D


def _raft():
    from ..extractors.raft import ExtractRAFT
    return ExtractRAFT


def _pwc():
    from ..extractors.pwc import ExtractPWC
    return ExtractPWC


def _vggish():
    from ..extractors.vggish import ExtractVGGish
    return ExtractVGGish


_REGISTRY: Dict[str, Callable] = {
    'CLIP-ViT-B/32': _clip,
    'CLIP-ViT-B/16': _clip,
    'CLIP4CLIP-ViT-B-32': _clip,
    'CLIP-RN50': _clip,
    'CLIP-RN101': _clip,
    'resnet18': _resnet,
    'resnet34': _resnet,
    'resnet50': _resnet,
    'resnet101': _resnet,
    'resnet152': _resnet,
    'r21d_rgb': _r21d,
    'i3d': _i3d,
    'raft': _raft,
    'pwc': _pwc,
    'vggish': _vggish,
    'vggish_torch': _vggish,
}


def get_extractor_class(feature_type: str):
    try:
        return _REGISTRY[feature_type]()
    except KeyError:
        raise ValueError(f'unknown feature_type {feature_type!r}; '
                         f'choices: {sorted(_REGISTRY)}') from None
