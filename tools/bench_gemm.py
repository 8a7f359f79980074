#!/usr/bin/env python3
"""Microbench: ops.linear_act (fused MFMA GEMM) vs torch F.linear(+act)
on the CLIP / VGGish / RAFT shapes.  Run on a GPU box."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import torch

from video_features_amd import ops

assert torch.cuda.is_available()
dev = torch.device('cuda:0')

SHAPES = [
    # (M, N, K, act, label)
    (9600, 3072, 768, 'quick_gelu', 'CLIP fc1 (fb192)'),
    (9600, 768, 3072, 'none', 'CLIP fc2'),
    (9600, 2304, 768, 'none', 'CLIP qkv'),
    (9600, 768, 768, 'none', 'CLIP proj'),
    (38400, 3072, 768, 'quick_gelu', 'CLIP fc1 (fb768)'),
    (4096, 4096, 12288, 'relu', 'VGGish fc1'),
    (200704, 256, 1920, 'none', 'RAFT GRU-zr-as-GEMM'),
    (4096, 4096, 4096, 'none', 'ladder 4096^3'),
    (8192, 8192, 8192, 'none', 'ladder 8192^3'),
    (19200, 3072, 768, 'quick_gelu', 'CLIP fc1 (fb384)'),
    (19200, 2304, 768, 'none', 'CLIP qkv (fb384)'),
    (1204224, 64, 64, 'relu', 'rn50 l1 conv1 (thin)'),
    (1204224, 256, 64, 'relu', 'rn50 l1 conv3 (thin)'),
    (1204224, 128, 256, 'relu', 'rn50 l2 conv1 (thin)'),
    (301056, 512, 128, 'relu', 'rn50 l2 conv3 (thin)'),
    (301056, 256, 512, 'relu', 'rn50 l3 conv1 (not thin)'),
]


def timeit(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


for m, n, k, act, label in SHAPES:
    torch.manual_seed(0)
    # VFA_GEMM_COLD=N: rotate N input buffers so A can't stay resident in
    # the 256 MiB Infinity Cache between iterations (in-model conditions —
    # cdna_hip_programming.md §2 L3 over-fetch masking); default 1 = warm
    ncold = max(1, int(os.environ.get('VFA_GEMM_COLD', '1')))
    xs = [(torch.randn(m, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
          for _ in range(ncold)]
    x = xs[0]
    w = (torch.randn(n, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    it = [0]

    def torch_path():
        it[0] += 1
        y = torch.nn.functional.linear(xs[it[0] % ncold], w, b)
        if act == 'relu':
            y = y.relu()
        elif act == 'quick_gelu':
            y = y * torch.sigmoid(1.702 * y)
        return y

    def ours():
        it[0] += 1
        return ops.linear_act(xs[it[0] % ncold], w, b, act)

    flops = 2.0 * m * n * k
    tt = timeit(torch_path)
    to = timeit(ours)
    # correctness spot-check (same buffer both paths)
    ref_t = torch.nn.functional.linear(x, w, b)
    if act == 'relu':
        ref_t = ref_t.relu()
    elif act == 'quick_gelu':
        ref_t = ref_t * torch.sigmoid(1.702 * ref_t)
    d = (ops.linear_act(x, w, b, act).float()
         - ref_t.float()).abs().max().item()
    ref = ref_t.float().abs().max().item()
    print(f'{label:<24} M{m:>7} N{n:>5} K{k:>6} {act:<11} '
          f'torch {tt * 1e6:7.1f}us ({flops / tt / 1e12:6.1f} TF) | '
          f'ours {to * 1e6:7.1f}us ({flops / to / 1e12:6.1f} TF) | '
          f'x{tt / to:4.2f} relerr {d / ref:.2e}', flush=True)
