#!/usr/bin/env python3
"""A/B bench: in-tree implicit-GEMM conv2d (conv2d.hip) vs MIOpen
(F.conv2d) on the hot conv shapes of the four model families.

Run on a GPU box:  python tools/bench_conv.py [--iters 50]
Prints per-shape TF/s for both paths and the ratio; >1.00 means the
in-tree kernel wins.  Evidence for profiles/ (VERDICT round-1 item 1).
"""
import argparse
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, '.')
from video_features_amd import ops  # noqa: E402

# (name, B, C, H, W, K, kh, kw, stride, pad)
SHAPES = [
    ('rn50 l1 3x3 64>64 56²',   384, 64, 56, 56, 64, 3, 3, 1, (1, 1)),
    ('rn50 l2 3x3 128>128 28²', 384, 128, 28, 28, 128, 3, 3, 1, (1, 1)),
    ('rn50 l3 3x3 256>256 14²', 384, 256, 14, 14, 256, 3, 3, 1, (1, 1)),
    ('rn50 l4 3x3 512>512 7²',  384, 512, 7, 7, 512, 3, 3, 1, (1, 1)),
    ('rn50 l2 s2 128 56²',      384, 128, 56, 56, 128, 3, 3, 2, (1, 1)),
    ('raft f 3x3 64>64 112²',    16, 64, 112, 112, 64, 3, 3, 1, (1, 1)),
    ('raft f 3x3 96>96 56²',     16, 96, 56, 56, 96, 3, 3, 1, (1, 1)),
    ('raft f 3x3 128>128 28²',   16, 128, 28, 28, 128, 3, 3, 1, (1, 1)),
    ('raft gru 1x5 384>256 28²', 64, 384, 28, 28, 256, 1, 5, 1, (0, 2)),
    ('raft gru 5x1 384>256 28²', 64, 384, 28, 28, 256, 5, 1, 1, (2, 0)),
    ('i3d 2c 3x3 64>576 56²',    96, 64, 56, 56, 576, 3, 3, 1, (1, 1)),
    ('i3d mix 3x3 96>384 28²',   96, 96, 28, 28, 384, 3, 3, 1, (1, 1)),
    ('vggish 3x3 64>128 48x32',  64, 64, 48, 32, 128, 3, 3, 1, (1, 1)),
    ('vggish 3x3 256>512 12x8',  64, 256, 12, 8, 512, 3, 3, 1, (1, 1)),
]


def bench(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters  # ms


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--iters', type=int, default=50)
    args = p.parse_args()
    assert ops.hip_available()
    dev = 'cuda:0'
    print(f"{'shape':28s} {'TFLOP':>6s} {'vfa ms':>8s} {'vfa TF/s':>9s} "
          f"{'miopen ms':>9s} {'mi TF/s':>8s} {'ratio':>6s}")
    wins = 0
    for name, b, c, h, w, k, kh, kw, st, pad in SHAPES:
        torch.manual_seed(0)
        x = (torch.randn(b, c, h, w, device=dev) * 0.5).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        wg = (torch.randn(k, c, kh, kw, device=dev) *
              (2.0 / (c * kh * kw)) ** 0.5).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        bias = torch.randn(k, device=dev).to(torch.bfloat16)
        oh = (h + 2 * pad[0] - kh) // st + 1
        ow = (w + 2 * pad[1] - kw) // st + 1
        flop = 2.0 * b * oh * ow * k * kh * kw * c

        t_vfa = bench(lambda: ops.conv2d_act(x, wg, bias, st, pad, 'relu'),
                      args.iters)
        t_mi = bench(
            lambda: F.relu(F.conv2d(x, wg, bias, st, pad)), args.iters)
        r = t_mi / t_vfa
        wins += r >= 1.0
        print(f'{name:28s} {flop / 1e12:6.2f} {t_vfa:8.3f} '
              f'{flop / t_vfa / 1e9:9.1f} {t_mi:9.3f} '
              f'{flop / t_mi / 1e9:8.1f} {r:6.2f}')
    print(f'\n{wins}/{len(SHAPES)} shapes >= MIOpen')


if __name__ == '__main__':
    main()
