"""Inference-time BatchNorm folding.

Every extractor runs eval-mode only (the reference never trains —
extraction is ``torch.no_grad()`` end to end), so each BatchNorm with
tracked running stats is an affine map that folds exactly into the
preceding conv:  w' = w * g/sqrt(v+eps),  b' = beta + (b - mean)*g/sqrt(v+eps).
On MI355X this removes one full HBM round-trip per conv (the
``batch_norm_transform_input`` kernel seen in profiles/).

InstanceNorm (RAFT fnet) computes per-sample statistics and cannot fold.
"""
from __future__ import annotations

import torch
from torch import nn

_BN = (nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d)
_CONV = (nn.Conv1d, nn.Conv2d, nn.Conv3d)

# (conv attr, bn attr) pairs used across this package's model families
_ATTR_PAIRS = [('conv', 'bn'), ('conv1', 'bn1'), ('conv2', 'bn2'),
               ('conv3', 'bn3'), ('conv1', 'norm1'), ('conv2', 'norm2'),
               ('spatial', 'bn')]


@torch.no_grad()
def fold_conv_bn(conv: nn.modules.conv._ConvNd, bn) -> None:
    """Fold ``bn`` (eval stats) into ``conv`` in place."""
    assert bn.track_running_stats and bn.running_var is not None
    g = (bn.weight if bn.weight is not None
         else torch.ones_like(bn.running_var))
    beta = (bn.bias if bn.bias is not None
            else torch.zeros_like(bn.running_mean))
    scale = g / torch.sqrt(bn.running_var + bn.eps)
    shape = [-1] + [1] * (conv.weight.dim() - 1)
    w = conv.weight.float() * scale.float().reshape(shape)
    b = torch.zeros_like(bn.running_mean) if conv.bias is None \
        else conv.bias.float()
    b = beta.float() + (b - bn.running_mean.float()) * scale.float()
    conv.weight.copy_(w.to(conv.weight.dtype))
    if conv.bias is None:
        conv.bias = nn.Parameter(b.to(conv.weight.dtype))
    else:
        conv.bias.copy_(b.to(conv.bias.dtype))


def _terminal_conv(m):
    """The conv whose output feeds the BN: the module itself, or the last
    conv of a composite block (R(2+1)D's Conv2Plus1D ends in .temporal)."""
    if isinstance(m, _CONV):
        return m
    tail = getattr(m, 'temporal', None)
    return tail if isinstance(tail, _CONV) else None


def fold_batchnorms(model: nn.Module) -> int:
    """Fold every foldable conv→BN pair in ``model`` (recursively), replacing
    the BN with Identity.  Returns the number of folds.  Model must be in
    eval mode."""
    assert not model.training, 'BN folding is inference-only'
    n = 0
    for m in model.modules():
        # known attribute pairs (Unit3D.conv/bn, ResidualBlock, ResNet
        # blocks, R(2+1)D Conv2Plus1D.spatial/bn, ...)
        for ca, ba in _ATTR_PAIRS:
            conv = _terminal_conv(getattr(m, ca, None))
            bn = getattr(m, ba, None)
            if conv is not None and isinstance(bn, _BN) \
                    and bn.track_running_stats:
                fold_conv_bn(conv, bn)
                setattr(m, ba, nn.Identity())
                n += 1
        # adjacent (conv, BN) inside Sequential containers
        if isinstance(m, nn.Sequential):
            for i in range(len(m) - 1):
                conv = _terminal_conv(m[i])
                if conv is not None and isinstance(m[i + 1], _BN) \
                        and m[i + 1].track_running_stats:
                    fold_conv_bn(conv, m[i + 1])
                    m[i + 1] = nn.Identity()
                    n += 1
    return n
