"""Native I3D (Inflated Inception-3D) for RGB and flow streams.

Re-implementation of the architecture the reference vendors
(reference models/i3d/i3d_src/i3d_net.py): Inception-v1 inflated to 3D with
TensorFlow-SAME padding semantics (asymmetric pads computed from the input
size), nine Mixed blocks, and a features mode that returns the 1024-d
embedding time-averaged over the clip
(reference i3d_net.py:238-264).

TF-SAME: for kernel k, stride s, input n the total pad is
``max(k - s, 0)`` when ``n % s == 0`` else ``max(k - n % s, 0)``, split with
the extra cell at the *end* — this differs from PyTorch's symmetric padding
and changes borders, so it is computed dynamically in forward via F.pad.
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops


def _same_pad_1d(n: int, k: int, s: int) -> Tuple[int, int]:
    total = max(k - s, 0) if n % s == 0 else max(k - (n % s), 0)
    front = total // 2
    return front, total - front


def tf_same_pad_3d(x: torch.Tensor, kernel, stride) -> torch.Tensor:
    """Pad (B, C, T, H, W) with TF-SAME semantics for a 3D conv/pool."""
    t, h, w = x.shape[-3:]
    pt = _same_pad_1d(t, kernel[0], stride[0])
    ph = _same_pad_1d(h, kernel[1], stride[1])
    pw = _same_pad_1d(w, kernel[2], stride[2])
    # F.pad order: (w_lo, w_hi, h_lo, h_hi, t_lo, t_hi)
    return F.pad(x, (pw[0], pw[1], ph[0], ph[1], pt[0], pt[1]))


def _triple(v) -> Tuple[int, int, int]:
    return (v, v, v) if isinstance(v, int) else tuple(v)


# ----------------------------------------------------- flattened-time path
# See models/_flat3d.py: the backbone below the stem runs on (B*T, C, H, W)
# channels_last tensors (no conv3d, no Im3d2Col).
from ._flat3d import (flatten_time, unflatten_time, temporal_merge,
                      temporal_max, cached_cl_weight)


def temporal_merge3(y: torch.Tensor, b: int) -> torch.Tensor:
    """3-tap stride-1 merge (3x3x3 conv, TF-SAME temporal pad (1,1))."""
    return temporal_merge(y, b, kt=3, st=1, p0=1, bias_tap=1)


class Unit3D(nn.Module):
    """Conv3d + BN + ReLU with TF-SAME padding
    (the reference's ``Unit3Dpy``, i3d_net.py:37-105)."""

    def __init__(self, in_ch: int, out_ch: int, kernel=1, stride=1,
                 use_bn: bool = True, activation: bool = True,
                 use_bias: bool = False):
        super().__init__()
        self.kernel = _triple(kernel)
        self.stride = _triple(stride)
        # stride-1 + odd kernel ⇒ TF-SAME total pad k-1 splits symmetrically,
        # so the conv's own padding is exact and the F.pad copy is skipped
        # (true for every I3D conv except the 7×7×7/2 stem)
        self.static_same = all(s == 1 for s in self.stride) and \
            all(k % 2 == 1 for k in self.kernel)
        pad = tuple(k // 2 for k in self.kernel) if self.static_same else 0
        self.conv = nn.Conv3d(in_ch, out_ch, self.kernel, self.stride,
                              padding=pad, bias=use_bias)
        # torch-default eps (1e-5): the reference's Unit3Dpy uses
        # BatchNorm3d defaults (i3d_net.py:92), and published i3d_rgb.pt /
        # i3d_flow.pt checkpoints are calibrated against that — 1e-3 here
        # cost ~1e-3 absolute feature divergence (caught by
        # tests/test_reference_parity.py)
        self.bn = nn.BatchNorm3d(out_ch) if use_bn else None
        self.activation = activation

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.static_same:
            x = tf_same_pad_3d(x, self.kernel, self.stride)
        x = self.conv(x)
        if self.bn is not None:
            x = self.bn(x)
        return F.relu(x, inplace=True) if self.activation else x

    def forward_flat(self, xf: torch.Tensor, b: int) -> torch.Tensor:
        """Flattened-time forward on (B*T, C, H, W) channels_last (see the
        module comment): 1x1x1 → conv2d GEMM; 3x3x3 → one merged conv2d
        (3*O outputs) + temporal shift-add.  Only valid for the stride-1
        units (everything but the stem)."""
        kt = self.kernel[0]
        w5 = self.conv.weight
        bias = self.conv.bias
        if kt == 1:
            w2 = cached_cl_weight(self, 'w2', w5, lambda: w5[:, :, 0])
            if (self.kernel[1] == 1 and xf.is_cuda
                    and xf.dtype == torch.bfloat16
                    and not isinstance(self.bn, nn.BatchNorm3d)
                    and w5.shape[1] % 8 == 0 and ops.hip_available()
                    and xf.is_contiguous(memory_format=torch.channels_last)):
                # folded 1x1x1 conv (+bias+ReLU) as ONE fused MFMA GEMM on
                # the CL view: removes MIOpen's SubTensor zero-fill, the
                # bias pass, and the separate ReLU round trip
                return ops.conv1x1_act(
                    xf, w2, bias, 'relu' if self.activation else 'none')
            x = ops.conv2d_act(xf, w2, bias,
                               1, (self.kernel[1] // 2, self.kernel[2] // 2))
        else:  # 3x3x3, stride 1
            o = w5.shape[0]
            wcat = cached_cl_weight(
                self, 'wcat', w5,
                lambda: w5.permute(2, 0, 1, 3, 4).reshape(
                    3 * o, w5.shape[1], self.kernel[1], self.kernel[2]))
            if bias is not None:
                def mk_bcat():
                    bc = torch.zeros(3 * o, device=bias.device,
                                     dtype=bias.dtype)
                    bc[o:2 * o] = bias
                    return bc
                bcat = cached_cl_weight(self, 'bcat', bias, mk_bcat)
            else:
                bcat = None
            # merged-tap 3x3 conv (3*O outputs) through the in-tree
            # implicit-GEMM kernel (1.6-1.9x MIOpen on the I3D shapes,
            # gpurun_out/bench_conv_r2b.log)
            y = ops.conv2d_act(xf, wcat, bcat,
                               1, (self.kernel[1] // 2, self.kernel[2] // 2))
            if self.activation and not isinstance(self.bn, nn.BatchNorm3d):
                # BN folded: ReLU rides the merge kernel's epilogue
                return temporal_merge(y, b, kt=3, st=1, p0=1, bias_tap=1,
                                      relu=True)
            x = temporal_merge3(y, b)
        if isinstance(self.bn, nn.BatchNorm3d):
            # BatchNorm3d == BatchNorm2d per channel on the flattened view
            bn = self.bn
            x = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight,
                             bn.bias, bn.training, bn.momentum, bn.eps)
        return F.relu(x, inplace=True) if self.activation else x


class MaxPool3dSame(nn.Module):
    def __init__(self, kernel, stride):
        super().__init__()
        self.kernel = _triple(kernel)
        self.stride = _triple(stride)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # GPU: fused TF-SAME pool (no padded copy, no argmax indices);
        # CPU fallback inside the op is the F.pad + max_pool3d reference
        return ops.maxpool3d_same(x, self.kernel, self.stride)

    def forward_flat(self, xf: torch.Tensor, b: int) -> torch.Tensor:
        # separable window: spatial TF-SAME 2D pool, then shifted temporal
        # maximum (exact — max commutes over the separable window)
        nhwc = xf.is_contiguous(memory_format=torch.channels_last) \
            and not xf.is_contiguous()
        y = ops.maxpool2d_same(xf, self.kernel[1:], self.stride[1:],
                               nhwc=nhwc)
        if self.kernel[0] > 1 or self.stride[0] > 1:
            t = y.shape[0] // b
            p0, p1 = _same_pad_1d(t, self.kernel[0], self.stride[0])
            y = temporal_max(y, b, self.kernel[0], self.stride[0], p0, p1)
        return y


class Mixed(nn.Module):
    """Inception block: 1×1 | 1×1→3×3 | 1×1→3×3 | pool→1×1
    (reference i3d_net.py:123-157)."""

    def __init__(self, in_ch: int, out: List[int]):
        super().__init__()
        self.b0 = Unit3D(in_ch, out[0], 1)
        self.b1 = nn.Sequential(Unit3D(in_ch, out[1], 1),
                                Unit3D(out[1], out[2], 3))
        self.b2 = nn.Sequential(Unit3D(in_ch, out[3], 1),
                                Unit3D(out[3], out[4], 3))
        self.b3 = nn.Sequential(MaxPool3dSame(3, 1),
                                Unit3D(in_ch, out[5], 1))
        self.out_channels = out[0] + out[2] + out[4] + out[5]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.cat([self.b0(x), self.b1(x), self.b2(x), self.b3(x)], 1)

    def forward_flat(self, xf: torch.Tensor, b: int) -> torch.Tensor:
        outs = [self.b0.forward_flat(xf, b)]
        for branch in (self.b1, self.b2, self.b3):
            y = xf
            for m in branch:
                y = m.forward_flat(y, b)
            outs.append(y)
        return torch.cat(outs, 1)


class I3D(nn.Module):
    FEAT_DIM = 1024

    def __init__(self, num_classes: int = 400, modality: str = 'rgb',
                 dropout_prob: float = 0.0):
        super().__init__()
        in_ch = {'rgb': 3, 'flow': 2}[modality]
        self.modality = modality
        self.conv3d_1a_7x7 = Unit3D(in_ch, 64, 7, 2)
        self.maxPool3d_2a_3x3 = MaxPool3dSame((1, 3, 3), (1, 2, 2))
        self.conv3d_2b_1x1 = Unit3D(64, 64, 1)
        self.conv3d_2c_3x3 = Unit3D(64, 192, 3)
        self.maxPool3d_3a_3x3 = MaxPool3dSame((1, 3, 3), (1, 2, 2))
        self.mixed_3b = Mixed(192, [64, 96, 128, 16, 32, 32])      # → 256
        self.mixed_3c = Mixed(256, [128, 128, 192, 32, 96, 64])    # → 480
        self.maxPool3d_4a_3x3 = MaxPool3dSame(3, 2)
        self.mixed_4b = Mixed(480, [192, 96, 208, 16, 48, 64])     # → 512
        self.mixed_4c = Mixed(512, [160, 112, 224, 24, 64, 64])    # → 512
        self.mixed_4d = Mixed(512, [128, 128, 256, 24, 64, 64])    # → 512
        self.mixed_4e = Mixed(512, [112, 144, 288, 32, 64, 64])    # → 528
        self.mixed_4f = Mixed(528, [256, 160, 320, 32, 128, 128])  # → 832
        self.maxPool3d_5a_2x2 = MaxPool3dSame(2, 2)
        self.mixed_5b = Mixed(832, [256, 160, 320, 32, 128, 128])  # → 832
        self.mixed_5c = Mixed(832, [384, 192, 384, 48, 128, 128])  # → 1024
        self.dropout = nn.Dropout(dropout_prob)
        self.conv3d_0c_1x1 = Unit3D(1024, num_classes, 1, use_bn=False,
                                    activation=False, use_bias=True)

    def _backbone(self, x: torch.Tensor) -> torch.Tensor:
        b = x.shape[0]
        stem = self.conv3d_1a_7x7
        if (x.is_cuda and x.dtype == torch.bfloat16 and ops.hip_available()
                and not ops._env_flag('VFA_NO_CONV')):
            # stem 7x7x7/2 decomposed like the 3x3x3 units: flatten time,
            # ONE merged 7-tap conv2d (channel-padded C=3/2 -> 8, through
            # the in-tree implicit-GEMM kernel) + strided temporal
            # shift-add — the last real conv3d (MIOpen/CK) is gone
            t, hh, ww = x.shape[2:]
            xf = flatten_time(x)
            w5 = stem.conv.weight              # (64, C, 7, 7, 7)
            cin = w5.shape[1]
            c8 = (cin + 7) // 8 * 8
            wcat = cached_cl_weight(
                stem, 'stem_wcat', w5,
                lambda: F.pad(
                    w5.permute(2, 0, 1, 3, 4).reshape(7 * 64, cin, 7, 7),
                    (0, 0, 0, 0, 0, c8 - cin)))
            bias = stem.conv.bias
            if bias is not None:
                def mk_b():
                    z = torch.zeros(7 * 64, device=bias.device,
                                    dtype=bias.dtype)
                    # tap 3 is valid at every output position for the
                    # TF-SAME (front 2/3) stride-2 temporal pad
                    z[3 * 64:4 * 64] = bias
                    return z
                bcat = cached_cl_weight(stem, 'stem_bcat', bias, mk_b)
            else:
                bcat = None
            pt = _same_pad_1d(t, 7, 2)
            ph = _same_pad_1d(hh, 7, 2)
            pw = _same_pad_1d(ww, 7, 2)
            y = ops.conv2d_act(xf, wcat, bcat, 2,
                               (ph[0], ph[1], pw[0], pw[1]))
            bn_folded = not isinstance(stem.bn, nn.BatchNorm3d)
            xf = temporal_merge(y, b, kt=7, st=2, p0=pt[0], bias_tap=3,
                                relu=bn_folded, p1=pt[1])
            if not bn_folded:
                bn = stem.bn
                xf = F.relu(
                    F.batch_norm(xf, bn.running_mean, bn.running_var,
                                 bn.weight, bn.bias, False, bn.momentum,
                                 bn.eps), inplace=True)
        else:
            x = self.conv3d_1a_7x7(x)          # CPU path: real conv3d
            xf = flatten_time(x)               # (B*T, C, H, W) CL
        xf = self.maxPool3d_2a_3x3.forward_flat(xf, b)
        xf = self.conv3d_2b_1x1.forward_flat(xf, b)
        xf = self.conv3d_2c_3x3.forward_flat(xf, b)
        xf = self.maxPool3d_3a_3x3.forward_flat(xf, b)
        xf = self.mixed_3c.forward_flat(self.mixed_3b.forward_flat(xf, b), b)
        xf = self.maxPool3d_4a_3x3.forward_flat(xf, b)
        for m in (self.mixed_4b, self.mixed_4c, self.mixed_4d, self.mixed_4e,
                  self.mixed_4f):
            xf = m.forward_flat(xf, b)
        xf = self.maxPool3d_5a_2x2.forward_flat(xf, b)
        xf = self.mixed_5c.forward_flat(self.mixed_5b.forward_flat(xf, b), b)
        x = unflatten_time(xf, b)
        # avg pool (2, 7, 7), stride 1 (reference i3d_net.py:229-235)
        kt = min(2, x.shape[2])
        return F.avg_pool3d(x, (kt, min(7, x.shape[3]), min(7, x.shape[4])))

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        """(B, C, T, H, W) → (B, 1024): spatial squeeze + mean over remaining
        time (reference i3d_net.py:238-264, features=True)."""
        x = self._backbone(x)
        return x.mean(dim=(2, 3, 4))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.dropout(self._backbone(x))
        # the 1x1x1 logits conv over (B, 1024, t, 1, 1) as a plain linear
        # on the channel dim — keeps the model conv3d-free end to end
        u = self.conv3d_0c_1x1
        w = u.conv.weight.reshape(u.conv.weight.shape[0], -1)   # (K, 1024)
        logits = F.linear(x.squeeze(-1).squeeze(-1).transpose(1, 2), w,
                          u.conv.bias)           # (B, t, K)
        return logits.mean(dim=1)                # time-averaged class scores
