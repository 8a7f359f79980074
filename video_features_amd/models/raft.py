"""Native RAFT optical flow (Teed & Deng, ECCV'20 architecture).

Re-implementation of the network the reference vendors
(reference models/raft/raft_src/{raft,extractor,update,corr}.py):
stride-8 feature/context encoders, all-pairs 4-level correlation pyramid,
iterative SepConvGRU updates, convex-combination 8× upsampling.

MI355X mapping: the all-pairs correlation is a plain batched GEMM
(rocBLAS via torch.matmul), while the per-iteration pyramid *lookup* — a
bilinear gather of 4×(2r+1)² taps per pixel — dispatches through
``ops.grid_sample_bilinear`` (hand-written HIP gather kernel on GPU).
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops


# --------------------------------------------------------------- encoders
class ResidualBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, norm: str, stride: int = 1):
        super().__init__()

        def make_norm(c):
            if norm == 'instance':
                return nn.InstanceNorm2d(c)
            if norm == 'batch':
                return nn.BatchNorm2d(c)
            return nn.Identity()

        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, stride, 1)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, 1, 1)
        self.norm1 = make_norm(out_ch)
        self.norm2 = make_norm(out_ch)
        self.relu = nn.ReLU(inplace=True)
        if stride == 1 and in_ch == out_ch:
            self.downsample = None
        else:
            self.downsample = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride), make_norm(out_ch))

    def forward(self, x):
        y = self.relu(self.norm1(self.conv1(x)))
        y = self.relu(self.norm2(self.conv2(y)))
        identity = x if self.downsample is None else self.downsample(x)
        return self.relu(identity + y)


class BasicEncoder(nn.Module):
    """Stride-8 ResNet encoder (reference extractor.py:118-192)."""

    def __init__(self, output_dim: int = 256, norm: str = 'instance'):
        super().__init__()
        if norm == 'instance':
            self.norm1 = nn.InstanceNorm2d(64)
        elif norm == 'batch':
            self.norm1 = nn.BatchNorm2d(64)
        else:
            self.norm1 = nn.Identity()
        self.conv1 = nn.Conv2d(3, 64, 7, 2, 3)
        self.relu1 = nn.ReLU(inplace=True)
        self.layer1 = nn.Sequential(ResidualBlock(64, 64, norm),
                                    ResidualBlock(64, 64, norm))
        self.layer2 = nn.Sequential(ResidualBlock(64, 96, norm, 2),
                                    ResidualBlock(96, 96, norm))
        self.layer3 = nn.Sequential(ResidualBlock(96, 128, norm, 2),
                                    ResidualBlock(128, 128, norm))
        self.conv2 = nn.Conv2d(128, output_dim, 1)

    def forward(self, x):
        x = self.relu1(self.norm1(self.conv1(x)))
        x = self.layer3(self.layer2(self.layer1(x)))
        return self.conv2(x)


# ------------------------------------------------------------ correlation
class CorrPyramid:
    """All-pairs correlation + 4-level avg-pool pyramid + windowed lookup
    (reference corr.py:25-60)."""

    def __init__(self, fmap1: torch.Tensor, fmap2: torch.Tensor,
                 num_levels: int = 4, radius: int = 4):
        self.num_levels = num_levels
        self.radius = radius
        b, d, h, w = fmap1.shape
        f1 = fmap1.flatten(2).transpose(1, 2)        # (B, HW, D)
        f2 = fmap2.flatten(2)                        # (B, D, HW)
        corr = torch.matmul(f1, f2) / (d ** 0.5)     # rocBLAS batched GEMM
        corr = corr.reshape(b * h * w, 1, h, w)
        self.shape = (b, h, w)
        self.pyramid: List[torch.Tensor] = [corr]
        for _ in range(num_levels - 1):
            corr = F.avg_pool2d(corr, 2, 2)
            self.pyramid.append(corr)

    def __call__(self, coords: torch.Tensor) -> torch.Tensor:
        """coords (B, 2, H, W) in pixels at 1/8 res → (B, L*(2r+1)^2, H, W)."""
        r = self.radius
        b, h, w = self.shape
        coords = coords.permute(0, 2, 3, 1)          # (B, H, W, 2)
        out = []
        for lvl, corr in enumerate(self.pyramid):
            dx = torch.linspace(-r, r, 2 * r + 1, device=coords.device,
                                dtype=coords.dtype)
            delta = torch.stack(torch.meshgrid(dx, dx, indexing='ij'),
                                dim=-1).flip(-1)     # (2r+1, 2r+1, 2) xy order
            centroid = coords.reshape(b * h * w, 1, 1, 2) / (2 ** lvl)
            window = centroid + delta[None]
            sampled = ops.grid_sample_bilinear(corr, window)
            out.append(sampled.reshape(b, h, w, -1))
        return torch.cat(out, dim=-1).permute(0, 3, 1, 2).contiguous()


# ----------------------------------------------------------------- update
class FlowHead(nn.Module):
    def __init__(self, in_dim: int = 128, hidden: int = 256):
        super().__init__()
        self.conv1 = nn.Conv2d(in_dim, hidden, 3, 1, 1)
        self.conv2 = nn.Conv2d(hidden, 2, 3, 1, 1)

    def forward(self, x):
        return self.conv2(F.relu(self.conv1(x)))


class SepConvGRU(nn.Module):
    """Separable 1×5 / 5×1 ConvGRU (reference update.py:37-64)."""

    def __init__(self, hidden: int = 128, in_dim: int = 256):
        super().__init__()
        c = hidden + in_dim
        self.convz1 = nn.Conv2d(c, hidden, (1, 5), padding=(0, 2))
        self.convr1 = nn.Conv2d(c, hidden, (1, 5), padding=(0, 2))
        self.convq1 = nn.Conv2d(c, hidden, (1, 5), padding=(0, 2))
        self.convz2 = nn.Conv2d(c, hidden, (5, 1), padding=(2, 0))
        self.convr2 = nn.Conv2d(c, hidden, (5, 1), padding=(2, 0))
        self.convq2 = nn.Conv2d(c, hidden, (5, 1), padding=(2, 0))

    def _step(self, h, x, convz, convr, convq):
        hx = torch.cat([h, x], dim=1)
        z = torch.sigmoid(convz(hx))
        r = torch.sigmoid(convr(hx))
        q = torch.tanh(convq(torch.cat([r * h, x], dim=1)))
        return (1 - z) * h + z * q

    def forward(self, h, x):
        h = self._step(h, x, self.convz1, self.convr1, self.convq1)
        h = self._step(h, x, self.convz2, self.convr2, self.convq2)
        return h


class BasicMotionEncoder(nn.Module):
    """(corr, flow) → 128-d motion features (reference update.py:83-101)."""

    def __init__(self, corr_levels: int = 4, corr_radius: int = 4):
        super().__init__()
        corr_planes = corr_levels * (2 * corr_radius + 1) ** 2
        self.convc1 = nn.Conv2d(corr_planes, 256, 1)
        self.convc2 = nn.Conv2d(256, 192, 3, 1, 1)
        self.convf1 = nn.Conv2d(2, 128, 7, 1, 3)
        self.convf2 = nn.Conv2d(128, 64, 3, 1, 1)
        self.conv = nn.Conv2d(192 + 64, 128 - 2, 3, 1, 1)

    def forward(self, flow, corr):
        c = F.relu(self.convc2(F.relu(self.convc1(corr))))
        f = F.relu(self.convf2(F.relu(self.convf1(flow))))
        out = F.relu(self.conv(torch.cat([c, f], dim=1)))
        return torch.cat([out, flow], dim=1)


class BasicUpdateBlock(nn.Module):
    def __init__(self, hidden: int = 128, context: int = 128):
        super().__init__()
        self.encoder = BasicMotionEncoder()
        self.gru = SepConvGRU(hidden, in_dim=128 + context)
        self.flow_head = FlowHead(hidden, 256)
        self.mask = nn.Sequential(nn.Conv2d(hidden, 256, 3, 1, 1),
                                  nn.ReLU(inplace=True),
                                  nn.Conv2d(256, 64 * 9, 1))

    def forward(self, net, inp, corr, flow):
        motion = self.encoder(flow, corr)
        net = self.gru(net, torch.cat([inp, motion], dim=1))
        delta_flow = self.flow_head(net)
        up_mask = 0.25 * self.mask(net)
        return net, up_mask, delta_flow


# ------------------------------------------------------------------- RAFT
class RAFT(nn.Module):
    def __init__(self, hidden_dim: int = 128, context_dim: int = 128,
                 iters: int = 20):
        super().__init__()
        self.hdim, self.cdim = hidden_dim, context_dim
        self.iters = iters
        self.fnet = BasicEncoder(256, 'instance')
        self.cnet = BasicEncoder(hidden_dim + context_dim, 'batch')
        self.update_block = BasicUpdateBlock(hidden_dim, context_dim)

    @staticmethod
    def coords_grid(b: int, h: int, w: int, device, dtype):
        yy, xx = torch.meshgrid(torch.arange(h, device=device, dtype=dtype),
                                torch.arange(w, device=device, dtype=dtype),
                                indexing='ij')
        return torch.stack([xx, yy])[None].expand(b, -1, -1, -1).contiguous()

    @staticmethod
    def upsample_flow(flow: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        """Convex-combination 8× upsample (reference raft.py:100-111)."""
        b, _, h, w = flow.shape
        mask = mask.view(b, 1, 9, 8, 8, h, w).softmax(dim=2)
        up = F.unfold(8 * flow, 3, padding=1).view(b, 2, 9, 1, 1, h, w)
        up = (mask * up).sum(dim=2)                 # (B, 2, 8, 8, H, W)
        return up.permute(0, 1, 4, 2, 5, 3).reshape(b, 2, 8 * h, 8 * w)

    def forward(self, image1: torch.Tensor, image2: torch.Tensor,
                iters: int = None, test_mode: bool = True):
        """uint8-range (B, 3, H, W) pairs (H, W divisible by 8) → (B, 2, H, W)
        flow (reference raft.py:113-174)."""
        iters = iters or self.iters
        image1 = 2 * (image1 / 255.0) - 1.0
        image2 = 2 * (image2 / 255.0) - 1.0
        fmap1 = self.fnet(image1)
        fmap2 = self.fnet(image2)
        corr_fn = CorrPyramid(fmap1.float(), fmap2.float())
        cnet = self.cnet(image1)
        net, inp = torch.split(cnet, [self.hdim, self.cdim], dim=1)
        net = torch.tanh(net)
        inp = F.relu(inp)
        b, _, h8, w8 = fmap1.shape
        coords0 = self.coords_grid(b, h8, w8, fmap1.device, torch.float32)
        coords1 = coords0.clone()
        flow_up = None
        for _ in range(iters):
            corr = corr_fn(coords1).to(net.dtype)
            flow = (coords1 - coords0).to(net.dtype)
            net, up_mask, delta = self.update_block(net, inp, corr, flow)
            coords1 = coords1 + delta.float()
            flow_up = self.upsample_flow((coords1 - coords0).to(net.dtype),
                                         up_mask)
        if test_mode:
            return flow_up
        return coords1 - coords0, flow_up


class InputPadder:
    """Pad to a multiple of 8 (reference raft.py:27-44 semantics)."""

    def __init__(self, shape, mode: str = 'sintel'):
        h, w = shape[-2:]
        ph = ((h + 7) // 8) * 8 - h
        pw = ((w + 7) // 8) * 8 - w
        if mode == 'sintel':
            self._pad = [pw // 2, pw - pw // 2, ph // 2, ph - ph // 2]
        else:
            self._pad = [pw // 2, pw - pw // 2, 0, ph]

    def pad(self, *inputs):
        return [F.pad(x, self._pad, mode='replicate') for x in inputs]

    def unpad(self, x):
        h, w = x.shape[-2:]
        c = [self._pad[2], h - self._pad[3], self._pad[0], w - self._pad[1]]
        return x[..., c[0]:c[1], c[2]:c[3]]
