#!/usr/bin/env python3
"""End-to-end extractor benchmark on real (synthetic) video FILES: decode →
sample → preprocess → GPU model → features, with the per-stage profiler.
Complements bench.py (which isolates the GPU pipeline on synthetic tensors).

Usage: python tools/bench_extractor.py [--feature_type CLIP-ViT-B/32]
           [--videos 8] [--frames 120] [--size 224]
"""
import argparse
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

import numpy as np
import torch

from video_features_amd.config import Config
from video_features_amd.io.y4m import write_y4m
from video_features_amd.models.registry import get_extractor_class


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--feature_type', default='CLIP-ViT-B/32')
    p.add_argument('--videos', type=int, default=8)
    p.add_argument('--frames', type=int, default=120)
    p.add_argument('--size', type=int, default=224)
    p.add_argument('--extract_method', default='uni_12')
    p.add_argument('--batch_size', type=int, default=32)
    p.add_argument('--cpu', action='store_true')
    args = p.parse_args()

    tmp = tempfile.mkdtemp()
    rng = np.random.default_rng(0)
    paths = []
    for i in range(args.videos):
        frames = rng.integers(0, 256, (args.frames, args.size, args.size, 3),
                              dtype=np.uint8)
        path = os.path.join(tmp, f'v{i}.y4m')
        write_y4m(path, frames, fps=25.0)
        paths.append(path)

    dev = torch.device('cpu' if args.cpu or not torch.cuda.is_available()
                       else 'cuda:0')
    cfg = Config(feature_type=args.feature_type,
                 batch_size=args.batch_size, video_paths=paths,
                 cpu=dev.type == 'cpu', extract_method=args.extract_method,
                 profile=True, tmp_path=os.path.join(tmp, 't'))
    ex = get_extractor_class(cfg.feature_type)(cfg, external_call=True)
    ex.models_for(dev)                       # build outside the timing
    idxs = torch.arange(len(paths), device=dev)
    ex(idxs[:1])                             # warm (find/caches)
    t0 = time.perf_counter()
    out = ex(idxs)
    dt = time.perf_counter() - t0
    n_feat = sum(o[cfg.feature_type].shape[0] for o in out)
    print(f'{args.feature_type}: {len(out)} videos, {n_feat} features in '
          f'{dt:.3f} s -> {len(out) / dt:.2f} videos/s, '
          f'{n_feat / dt:.1f} features/s (file decode included)')


if __name__ == '__main__':
    main()
