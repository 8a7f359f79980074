"""Native ResNet-18/34/50/101/152 (feature extractor + classifier head).

The reference pulls these from the torchvision model zoo (reference
models/resnet/extract_resnet.py:54-67).  From-scratch implementation of the
standard v1 architecture; ``forward_features`` returns the post-avgpool
embedding (512-d for 18/34, 2048-d for 50/101/152) and ``forward`` the
1000-way ImageNet logits used by ``--show_pred``.

Convolutions run through PyTorch-ROCm (MIOpen); on MI355X the surrounding
normalization/activation work is fused by channels_last memory format.
"""
from __future__ import annotations

from typing import List, Type

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, 1, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = None
        if stride != 1 or in_planes != planes:
            self.downsample = nn.Sequential(
                nn.Conv2d(in_planes, planes, 1, stride, bias=False),
                nn.BatchNorm2d(planes))

    def forward(self, x):
        if (isinstance(self.bn1, nn.Identity) and x.is_cuda
                and x.dtype == torch.bfloat16
                and x.is_contiguous(memory_format=torch.channels_last)
                and ops.hip_available()):
            # both 3x3 convs through the in-tree implicit-GEMM kernel;
            # conv2 fuses the residual add + ReLU into its epilogue.
            # The downsample 1x1 (BN folded -> [conv, Identity]) also goes
            # in-tree: nn.Conv2d.forward would hand it to MIOpen.
            idt = x if self.downsample is None else \
                ops.conv2d_mod(self.downsample[0], x, 'none')
            if not idt.is_contiguous(memory_format=torch.channels_last):
                idt = idt.contiguous(memory_format=torch.channels_last)
            y = ops.conv2d_act(x, self.conv1.weight, self.conv1.bias,
                               self.conv1.stride[0], 1, 'relu')
            return ops.conv2d_act(y, self.conv2.weight, self.conv2.bias,
                                  1, 1, 'relu', res=idt)
        identity = x if self.downsample is None else self.downsample(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        out_planes = planes * self.expansion
        self.conv1 = nn.Conv2d(in_planes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, out_planes, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_planes)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = None
        if stride != 1 or in_planes != out_planes:
            self.downsample = nn.Sequential(
                nn.Conv2d(in_planes, out_planes, 1, stride, bias=False),
                nn.BatchNorm2d(out_planes))

    def _fused_ready(self, x) -> bool:
        # inference path: BNs folded into the convs, bf16 channels_last on
        # GPU with the HIP extension present
        return (isinstance(self.bn1, nn.Identity) and x.is_cuda
                and x.dtype == torch.bfloat16
                and x.is_contiguous(memory_format=torch.channels_last)
                and ops.hip_available()
                and not ops._env_flag('VFA_NO_LTGEMM'))

    def forward(self, x):
        if self._fused_ready(x):
            # 1x1 convs as fused MFMA GEMMs on the CL view: conv1+ReLU in
            # one epilogue; conv3 + residual-add + ReLU in one epilogue
            # (removes MIOpen's SubTensor zero-fill + the eager add/relu
            # round trips — see profiles/).  The downsample 1x1 (stride 1
            # or 2; BN folded -> [conv, Identity]) routes in-tree too:
            # left as self.downsample(x) it was ~12% of steady-state
            # kernel time on MIOpen igemm + SubTensor zero-fill
            # (profiles/resnet50_profile_final_r02.md).
            idt = x if self.downsample is None else \
                ops.conv2d_mod(self.downsample[0], x, 'none')
            y = ops.conv1x1_act(x, self.conv1.weight, self.conv1.bias,
                                'relu')
            # 3x3 conv + bias + ReLU as ONE in-tree implicit-GEMM MFMA
            # kernel (conv2d.hip — 1.4-1.5x MIOpen on these shapes,
            # gpurun_out/bench_conv_r2b.log)
            y = ops.conv2d_act(y, self.conv2.weight, self.conv2.bias,
                               self.conv2.stride[0], 1, 'relu')
            if not idt.is_contiguous(memory_format=torch.channels_last):
                idt = idt.contiguous(memory_format=torch.channels_last)
            return ops.conv1x1_act(y, self.conv3.weight, self.conv3.bias,
                                   'relu', res=idt)
        identity = x if self.downsample is None else self.downsample(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, block: Type, layers: List[int], num_classes: int = 1000):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, 2, 3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        self.in_planes = 64
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.feat_dim = 512 * block.expansion
        self.fc = nn.Linear(self.feat_dim, num_classes)

    def _make_layer(self, block, planes, n, stride):
        blocks = [block(self.in_planes, planes, stride)]
        self.in_planes = planes * block.expansion
        blocks += [block(self.in_planes, planes) for _ in range(n - 1)]
        return nn.Sequential(*blocks)

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        # stem 7x7 (C=3) through the in-tree kernel via channel padding;
        # ReLU rides the conv epilogue once BN is folded
        bn1_id = isinstance(self.bn1, nn.Identity)
        x = ops.conv2d_mod(self.conv1, x, 'relu' if bn1_id else 'none')
        if not bn1_id:
            x = self.relu(self.bn1(x))
        if (x.is_cuda and ops.hip_available()
                and x.is_contiguous(memory_format=torch.channels_last)
                and ops._env_flag('VFA_STEMPOOL')):
            # opt-in: the in-tree pool kernel measured ~3% SLOWER end-to-end
            # than torch's NHWC maxpool at this shape (64ch 112x112; same-box
            # A/B 32.3k vs 33.3k f/s), so torch stays the default here
            x = ops.maxpool2d(x, (3, 3), (2, 2), 1, nhwc=True)
        else:
            x = self.maxpool(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.avgpool(x).flatten(1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc(self.forward_features(x))


_SPECS = {
    'resnet18': (BasicBlock, [2, 2, 2, 2]),
    'resnet34': (BasicBlock, [3, 4, 6, 3]),
    'resnet50': (Bottleneck, [3, 4, 6, 3]),
    'resnet101': (Bottleneck, [3, 4, 23, 3]),
    'resnet152': (Bottleneck, [3, 8, 36, 3]),
}


def build_resnet(name: str, num_classes: int = 1000) -> ResNet:
    block, layers = _SPECS[name]
    return ResNet(block, layers, num_classes)
