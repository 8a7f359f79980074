"""Native R(2+1)D-18 video network.

The reference uses ``torchvision.models.video.r2plus1d_18``
(reference models/r21d/extract_r21d.py:57).  From-scratch implementation of
the R(2+1)D factorization: every 3D conv t×k×k is a spatial (1,k,k) conv
into an intermediate width M, BN+ReLU, then a temporal (t,1,1) conv, with
M chosen so the parameter count matches the full 3D conv:
``M = (t*k*k*Cin*Cout) / (k*k*Cin + t*Cout)`` (Tran et al., CVPR'18).

``forward_features`` returns the 512-d post-avgpool embedding per clip;
``forward`` the 400-way Kinetics logits for ``--show_pred``.
"""
from __future__ import annotations

import torch
from torch import nn


def _midplanes(in_planes: int, out_planes: int, t: int = 3, k: int = 3) -> int:
    return (t * k * k * in_planes * out_planes) // (
        k * k * in_planes + t * out_planes)


class Conv2Plus1D(nn.Module):
    def __init__(self, in_planes: int, out_planes: int, mid: int,
                 stride=(1, 1, 1)):
        super().__init__()
        st, ss = stride[0], stride[1]
        self.spatial = nn.Conv3d(in_planes, mid, (1, 3, 3), (1, ss, ss),
                                 (0, 1, 1), bias=False)
        self.bn = nn.BatchNorm3d(mid)
        self.relu = nn.ReLU(inplace=True)
        self.temporal = nn.Conv3d(mid, out_planes, (3, 1, 1), (st, 1, 1),
                                  (1, 0, 0), bias=False)

    def forward(self, x):
        return self.temporal(self.relu(self.bn(self.spatial(x))))


class R21DBlock(nn.Module):
    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        s = (stride, stride, stride)
        # midplanes computed once per block from (in, out) and shared by both
        # convs — the published R(2+1)D-18 parameterization (31.5M params)
        mid = _midplanes(in_planes, planes)
        self.conv1 = Conv2Plus1D(in_planes, planes, mid, s)
        self.bn1 = nn.BatchNorm3d(planes)
        self.conv2 = Conv2Plus1D(planes, planes, mid)
        self.bn2 = nn.BatchNorm3d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = None
        if stride != 1 or in_planes != planes:
            self.downsample = nn.Sequential(
                nn.Conv3d(in_planes, planes, 1, s, bias=False),
                nn.BatchNorm3d(planes))

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + identity)


class R2Plus1D18(nn.Module):
    def __init__(self, num_classes: int = 400):
        super().__init__()
        # R(2+1)D stem: (1,7,7) spatial s(1,2,2) into 45 ch, then (3,1,1)
        self.stem = nn.Sequential(
            nn.Conv3d(3, 45, (1, 7, 7), (1, 2, 2), (0, 3, 3), bias=False),
            nn.BatchNorm3d(45), nn.ReLU(inplace=True),
            nn.Conv3d(45, 64, (3, 1, 1), (1, 1, 1), (1, 0, 0), bias=False),
            nn.BatchNorm3d(64), nn.ReLU(inplace=True))
        self.layer1 = nn.Sequential(R21DBlock(64, 64), R21DBlock(64, 64))
        self.layer2 = nn.Sequential(R21DBlock(64, 128, 2), R21DBlock(128, 128))
        self.layer3 = nn.Sequential(R21DBlock(128, 256, 2), R21DBlock(256, 256))
        self.layer4 = nn.Sequential(R21DBlock(256, 512, 2), R21DBlock(512, 512))
        self.avgpool = nn.AdaptiveAvgPool3d(1)
        self.feat_dim = 512
        self.fc = nn.Linear(512, num_classes)

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        """(B, 3, T, H, W) → (B, 512)."""
        x = self.stem(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.avgpool(x).flatten(1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc(self.forward_features(x))
