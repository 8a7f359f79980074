"""Native R(2+1)D video networks (depths 18 and 34).

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
import torch.nn.functional as F
from torch import nn

from .. import ops
from ._flat3d import (flatten_time, temporal_merge, temporal_select,
                      cached_cl_weight)


def _flat_bn_relu(x, bn, relu: bool):
    if isinstance(bn, nn.BatchNorm3d):
        x = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight,
                         bn.bias, bn.training, bn.momentum, bn.eps)
    return F.relu(x, inplace=True) if relu else x


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

    def forward_flat(self, xf, b, relu_after: bool = False):
        """(B*T, C, H, W) channels_last path (see models/_flat3d.py):
        spatial (1,3,3) conv = conv2d; temporal (3,1,1) conv = merged
        3-tap 1x1 conv2d + shifted (strided) temporal add (with the
        follow-up ReLU fused into the merge when requested)."""
        ss = self.spatial.stride[1]
        sw = cached_cl_weight(self, 'sw', self.spatial.weight,
                              lambda: self.spatial.weight[:, :, 0])
        y = ops.conv2d_act(xf, sw, self.spatial.bias, ss, 1)
        y = _flat_bn_relu(y, self.bn, True)
        w = self.temporal.weight                   # (O, M, 3, 1, 1)
        o, mid = w.shape[0], w.shape[1]
        # R(2+1)D mid-planes (144/230/460/921...) are often not %8; on GPU
        # the weight is zero-padded to c8 once (cached) and the conv
        # kernel's pad pass widens the input — in-tree instead of MIOpen
        c8 = ((mid + 7) // 8 * 8
              if w.is_cuda and w.dtype == torch.bfloat16 else mid)
        wcat = cached_cl_weight(
            self, 'wcat', w,
            lambda: F.pad(
                w.permute(2, 0, 1, 3, 4).reshape(3 * o, mid, 1, 1),
                (0, 0, 0, 0, 0, c8 - mid)))
        bias = self.temporal.bias
        if bias is not None:
            def mk_bcat():
                bc = torch.zeros(3 * o, device=bias.device, dtype=bias.dtype)
                bc[o:2 * o] = bias
                return bc
            bcat = cached_cl_weight(self, 'bcat', bias, mk_bcat)
        else:
            bcat = None
        y = ops.conv2d_act(y, wcat, bcat)
        return temporal_merge(y, b, kt=3, st=self.temporal.stride[0], p0=1,
                              relu=relu_after)


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

    def forward_flat(self, xf, b):
        if self.downsample is None:
            identity = xf
        else:
            conv, bn = self.downsample[0], self.downsample[1]
            st = conv.stride[0]
            # 1x1x1 stride-s conv: temporal subsample, then conv2d stride s
            identity = temporal_select(xf, b, st)
            dw = cached_cl_weight(self, 'dw', conv.weight,
                                  lambda: conv.weight[:, :, 0])
            identity = ops.conv2d_act(identity, dw, conv.bias,
                                      conv.stride[1], 0)
            identity = _flat_bn_relu(identity, bn, False)
        bn1_folded = isinstance(self.bn1, nn.Identity)
        out = self.conv1.forward_flat(xf, b, relu_after=bn1_folded)
        if not bn1_folded:
            out = _flat_bn_relu(out, self.bn1, True)
        out = self.conv2.forward_flat(out, b)
        out = _flat_bn_relu(out, self.bn2, False)
        return F.relu(out + identity, inplace=True)


class R2Plus1D18(nn.Module):
    """R(2+1)D with basic blocks.  ``layers`` gives the per-stage block
    counts: (2,2,2,2) is R(2+1)D-18 (torchvision's ``r2plus1d_18``,
    reference extract_r21d.py:57), (3,4,6,3) is R(2+1)D-34 (the IG-65M
    depth BASELINE.json config 5 names)."""

    def __init__(self, num_classes: int = 400, layers=(2, 2, 2, 2)):
        super().__init__()
        # R(2+1)D stem: (1,7,7) spatial s(1,2,2) into 45 ch, then (3,1,1)
        self.stem = nn.Sequential(
            nn.Conv3d(3, 45, (1, 7, 7), (1, 2, 2), (0, 3, 3), bias=False),
            nn.BatchNorm3d(45), nn.ReLU(inplace=True),
            nn.Conv3d(45, 64, (3, 1, 1), (1, 1, 1), (1, 0, 0), bias=False),
            nn.BatchNorm3d(64), nn.ReLU(inplace=True))

        def stage(in_p, out_p, n, stride):
            blocks = [R21DBlock(in_p, out_p, stride)]
            blocks += [R21DBlock(out_p, out_p) for _ in range(n - 1)]
            return nn.Sequential(*blocks)

        self.layer1 = stage(64, 64, layers[0], 1)
        self.layer2 = stage(64, 128, layers[1], 2)
        self.layer3 = stage(128, 256, layers[2], 2)
        self.layer4 = stage(256, 512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool3d(1)
        self.feat_dim = 512
        self.fc = nn.Linear(512, num_classes)

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        """(B, 3, T, H, W) → (B, 512) via the flattened-time path — the
        R(2+1)D factorization is fully separable, so NO conv3d runs at
        all (reference uses torchvision's NCDHW conv3d stack)."""
        b = x.shape[0]
        xf = flatten_time(x)
        # stem: (1,7,7)/s(1,2,2) conv2d -> BN+ReLU -> 3-tap temporal 1x1
        sc0, sbn0, sc1, sbn1 = (self.stem[0], self.stem[1], self.stem[3],
                                self.stem[4])
        stem_c8 = (8 if sc0.weight.is_cuda
                   and sc0.weight.dtype == torch.bfloat16 else 3)
        s0w = cached_cl_weight(
            self, 's0w', sc0.weight,
            lambda: F.pad(sc0.weight[:, :, 0], (0, 0, 0, 0, 0, stem_c8 - 3)))
        xf = ops.conv2d_act(xf, s0w, sc0.bias, 2, 3)
        xf = _flat_bn_relu(xf, sbn0, True)
        o, mid = sc1.weight.shape[0], sc1.weight.shape[1]
        c8 = ((mid + 7) // 8 * 8
              if sc1.weight.is_cuda and sc1.weight.dtype == torch.bfloat16
              else mid)
        wcat = cached_cl_weight(
            self, 's1w', sc1.weight,
            lambda: F.pad(
                sc1.weight.permute(2, 0, 1, 3, 4).reshape(3 * o, mid, 1, 1),
                (0, 0, 0, 0, 0, c8 - mid)))
        if sc1.bias is not None:
            def mk_bcat():
                bc = torch.zeros(3 * o, device=sc1.bias.device,
                                 dtype=sc1.bias.dtype)
                bc[o:2 * o] = sc1.bias
                return bc
            bcat = cached_cl_weight(self, 's1b', sc1.bias, mk_bcat)
        else:
            bcat = None
        sbn1_folded = isinstance(sbn1, nn.Identity)
        xf = temporal_merge(ops.conv2d_act(xf, wcat, bcat), b, kt=3, st=1,
                            p0=1, relu=sbn1_folded)
        if not sbn1_folded:
            xf = _flat_bn_relu(xf, sbn1, True)
        for layer in (self.layer1, self.layer2, self.layer3, self.layer4):
            for blk in layer:
                xf = blk.forward_flat(xf, b)
        # global average over (T, H, W)
        bt, c, h, w = xf.shape
        return xf.view(b, bt // b, c, h, w).mean(dim=(1, 3, 4))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc(self.forward_features(x))


def R2Plus1D34(num_classes: int = 400) -> R2Plus1D18:
    """R(2+1)D-34: basic blocks at depths (3, 4, 6, 3)."""
    return R2Plus1D18(num_classes, layers=(3, 4, 6, 3))
