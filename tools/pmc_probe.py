#!/usr/bin/env python3
"""Minimal kernel probe for rocprofv3 --pmc counter runs: exercises ONLY the
hand-written HIP kernels (no MIOpen find phase) a few times each."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import torch

from video_features_amd import ops

assert torch.cuda.is_available()
dev = torch.device('cuda:0')
torch.manual_seed(0)

# fused MFMA linear: CLIP fc1 shape (BIG tile) + fb192 (small tile)
x1 = (torch.randn(19200, 768, device=dev) / 5).to(torch.bfloat16)
w1 = (torch.randn(3072, 768, device=dev) / 5).to(torch.bfloat16)
b1 = torch.randn(3072, device=dev).to(torch.bfloat16)
x2 = x1[:9600].contiguous()

# flash attention: CLIP shape
qkv = torch.randn(384, 50, 3, 12, 64, device=dev).to(torch.bfloat16) / 3

# corr lookup: RAFT bench shape
pyr = [torch.randn(256 * 784, 1, 28 >> i, 28 >> i, device=dev)
       for i in range(4)]
coords = torch.rand(256, 2, 28, 28, device=dev) * 28

# layernorm-residual
ln_x = torch.randn(19200, 768, device=dev).to(torch.bfloat16)
ln_w = torch.ones(768, device=dev).to(torch.bfloat16)
ln_b = torch.zeros(768, device=dev).to(torch.bfloat16)

# implicit-GEMM conv2d: the ResNet layer2 3x3 (stride 1 + stride 2)
cx = (torch.randn(64, 128, 56, 56, device=dev) / 8).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
cw = (torch.randn(128, 128, 3, 3, device=dev) / 8).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
cb = torch.randn(128, device=dev).to(torch.bfloat16)

# thin-K streaming GEMM: ResNet layer1 1x1 shape (K=64) + ragged K=144
tx = (torch.randn(401408, 64, device=dev) / 5).to(torch.bfloat16)
tw = (torch.randn(256, 64, device=dev) / 5).to(torch.bfloat16)
tx2 = (torch.randn(100000, 144, device=dev) / 5).to(torch.bfloat16)
tw2 = (torch.randn(288, 144, device=dev) / 5).to(torch.bfloat16)

# deep-pipelined 8-phase GEMM (act-none route)
px = (torch.randn(8192, 768, device=dev) / 5).to(torch.bfloat16)
pw = (torch.randn(768, 768, device=dev) / 5).to(torch.bfloat16)

with torch.no_grad():
    for _ in range(5):
        ops.linear_act(x1, w1, b1, 'quick_gelu')
        ops.linear_act(x2, w1, b1, 'quick_gelu')
        ops.conv2d_act(cx, cw, cb, 1, 1, 'relu')
        ops.conv2d_act(cx, cw, cb, 2, 1, 'relu')
        ops.linear_act(tx, tw, None, 'relu')
        ops.linear_act(tx2, tw2, None, 'relu')
        ops.linear_act(px, pw, None, 'none')
        ops.mhsa_fused(qkv.reshape(384, 50, -1), 12)
        ops.corr_lookup(pyr, coords, 4, True, torch.bfloat16)
        ops.layer_norm_residual(ln_x, ln_x, ln_w, ln_b)
        ops.instance_norm(
            torch.randn(128, 64, 112, 112, device=dev).to(torch.bfloat16)
            .contiguous(memory_format=torch.channels_last),
            relu=True, nhwc=True)
torch.cuda.synchronize()
print('pmc probe done')
