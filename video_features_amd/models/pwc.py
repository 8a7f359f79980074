"""Native PWC-Net optical flow (Sun et al., CVPR'18 architecture).

Re-implementation of the network the reference vendors
(reference models/pwc/pwc_src/pwc_net.py): 6-level feature pyramid,
per-level cost volume over a 9×9 displacement window, DenseNet-style
decoders with backward warping, dilated-conv context refiner, ×20 flow
scaling with pad-to-/64 bilinear resizing.

MI355X mapping: the 81-channel cost volume — which the reference JIT-compiles
from CUDA strings via CuPy (reference models/pwc/pwc_src/correlation.py) —
dispatches through ``ops.pwc_correlation``: ONE fused hand-written CDNA4 HIP
kernel on GPU (no rearrange pass, LDS-staged center features, wave64
channel reduction), and a vectorized torch fallback on CPU.  Warping goes
through ``ops.bilinear_warp``.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops


class _ConvLeaky(nn.Sequential):
    """conv + LeakyReLU(0.1) with the Sequential key scheme (``0.weight``)
    preserved; on GPU the pair runs as ONE fused implicit-GEMM kernel
    (act=leaky_relu) via conv2d_mod — dilated refiner convs fall back."""

    def forward(self, x):
        return ops.conv2d_mod(self[0], x, 'leaky_relu')


def _conv(in_ch: int, out_ch: int, stride: int = 1, dilation: int = 1):
    return _ConvLeaky(
        nn.Conv2d(in_ch, out_ch, 3, stride, padding=dilation, dilation=dilation),
        nn.LeakyReLU(0.1, inplace=True))


class PyramidExtractor(nn.Module):
    """6-level siamese feature pyramid, channels 16/32/64/96/128/196
    (reference pwc_net.py:44-110)."""

    CHANNELS = [16, 32, 64, 96, 128, 196]

    def __init__(self):
        super().__init__()
        levels = []
        in_ch = 3
        for out_ch in self.CHANNELS:
            levels.append(nn.Sequential(_conv(in_ch, out_ch, 2),
                                        _conv(out_ch, out_ch),
                                        _conv(out_ch, out_ch)))
            in_ch = out_ch
        self.levels = nn.ModuleList(levels)

    def forward(self, x: torch.Tensor) -> List[torch.Tensor]:
        if x.is_cuda and x.dtype == torch.bfloat16:
            # channels_last through the pyramid so the fused conv kernel
            # runs; the decoder's correlation kernel wants NCHW, its
            # .contiguous() converts per level
            x = x.contiguous(memory_format=torch.channels_last)
        feats = []
        for level in self.levels:
            x = level(x)
            feats.append(x)
        return feats   # [1/2 .. 1/64]


class Decoder(nn.Module):
    """Per-level flow decoder with dense connections
    (reference pwc_net.py:113-186)."""

    DENSE = [128, 128, 96, 64, 32]

    def __init__(self, feat_ch: int, top: bool = False, last: bool = False):
        super().__init__()
        self.top = top
        in_ch = 81 if top else 81 + feat_ch + 2 + 2   # corr + f1 + upflow + upfeat
        convs = []
        c = in_ch
        for out_ch in self.DENSE:
            convs.append(_conv(c, out_ch))
            c += out_ch
        self.convs = nn.ModuleList(convs)
        self.predict = nn.Conv2d(c, 2, 3, 1, 1)
        self.out_channels = c
        # upflow/upfeat feed the NEXT (finer) level; the bottom decoder
        # (level 2) has no consumer, so the parameter set matches the
        # reference's exactly (its Decoder(l) holds the upsamplers the
        # consumer side, pwc_net.py:119-120 — same tensors, shifted owner)
        if not last:
            self.upflow = nn.ConvTranspose2d(2, 2, 4, 2, 1)
            self.upfeat = nn.ConvTranspose2d(c, 2, 4, 2, 1)

    def forward(self, f1: torch.Tensor, f2: torch.Tensor,
                upflow: Optional[torch.Tensor],
                upfeat: Optional[torch.Tensor],
                warp_scale: float) -> Tuple[torch.Tensor, torch.Tensor]:
        if self.top:
            corr = F.leaky_relu(ops.pwc_correlation(f1, f2), 0.1)
            x = corr
        else:
            warped = ops.bilinear_warp(f2, upflow * warp_scale)
            corr = F.leaky_relu(ops.pwc_correlation(f1, warped), 0.1)
            x = torch.cat([corr, f1, upflow, upfeat], dim=1)
        for conv in self.convs:
            x = torch.cat([conv(x), x], dim=1)
        flow = self.predict(x)
        return flow, x


class Refiner(nn.Module):
    """Dilated-conv context network (reference pwc_net.py:189-210)."""

    def __init__(self, in_ch: int):
        super().__init__()
        self.net = nn.Sequential(
            _conv(in_ch, 128, dilation=1), _conv(128, 128, dilation=2),
            _conv(128, 128, dilation=4), _conv(128, 96, dilation=8),
            _conv(96, 64, dilation=16), _conv(64, 32, dilation=1),
            nn.Conv2d(32, 2, 3, 1, 1))

    def forward(self, x):
        return self.net(x)


class PWCNet(nn.Module):
    # flow at level l is in level-l pixel units; upsampled flow must be scaled
    # before warping the next level's features (reference pwc_net.py:250-254)
    WARP_SCALES = {5: 0.625, 4: 1.25, 3: 2.5, 2: 5.0}

    def __init__(self):
        super().__init__()
        self.extractor = PyramidExtractor()
        ch = PyramidExtractor.CHANNELS       # [16, 32, 64, 96, 128, 196]
        self.decoder6 = Decoder(ch[5], top=True)
        self.decoder5 = Decoder(ch[4])
        self.decoder4 = Decoder(ch[3])
        self.decoder3 = Decoder(ch[2])
        self.decoder2 = Decoder(ch[1], last=True)
        self.refiner = Refiner(self.decoder2.out_channels)

    def forward(self, im1: torch.Tensor, im2: torch.Tensor) -> torch.Tensor:
        """uint8-range RGB (B, 3, H, W) pairs → (B, 2, H, W) flow in input
        pixel units (reference pwc_net.py:213-261: BGR/255 input, /64
        bilinear pad, ×20 flow scale, final bilinear upsample)."""
        b, _, h, w = im1.shape
        im1 = im1.flip(1) / 255.0            # RGB → BGR, [0, 1]
        im2 = im2.flip(1) / 255.0
        h64 = ((h + 63) // 64) * 64
        w64 = ((w + 63) // 64) * 64
        if (h64, w64) != (h, w):
            im1 = F.interpolate(im1, (h64, w64), mode='bilinear', align_corners=False)
            im2 = F.interpolate(im2, (h64, w64), mode='bilinear', align_corners=False)
        p1 = self.extractor(im1)
        p2 = self.extractor(im2)
        # pyramid list index: level l features are p[l-1] (1/2^l resolution)
        flow6, feat6 = self.decoder6(p1[5], p2[5], None, None, 0.0)
        flow, feat = flow6, feat6
        dec = {5: self.decoder5, 4: self.decoder4, 3: self.decoder3,
               2: self.decoder2}
        for lvl in (5, 4, 3, 2):
            prev_dec = {6: self.decoder6, 5: self.decoder5, 4: self.decoder4,
                        3: self.decoder3}[lvl + 1]
            upflow = prev_dec.upflow(flow)
            upfeat = prev_dec.upfeat(feat)
            flow, feat = dec[lvl](p1[lvl - 1], p2[lvl - 1], upflow, upfeat,
                                  self.WARP_SCALES[lvl])
        flow = flow + self.refiner(feat)
        # ×20 to pixel units at 1/4 res, then resize to the input resolution
        flow = F.interpolate(flow * 20.0, (h, w), mode='bilinear',
                             align_corners=False)
        flow[:, 0] *= float(w) / float(w64)
        flow[:, 1] *= float(h) / float(h64)
        return flow
