"""CLI — flag-compatible with the reference's ``main.py``
(reference main.py:93-149), plus a few additive knobs (``--dtype``,
``--resume``, ``--profile``, ``--gather_features``, ``--seed``,
``--weights_path``)."""
from __future__ import annotations

import argparse
from typing import List, Optional

from .config import FEATURE_TYPES, Config, sanity_check
from .runtime.dist import run_extraction


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description='Extract Features (MI355X-native)')
    p.add_argument('--feature_type', required=True, choices=FEATURE_TYPES)
    p.add_argument('--video_paths', nargs='+',
                   help='space-separated paths to videos')
    p.add_argument('--flow_paths', nargs='+',
                   help='space-separated paths to video flow images')
    p.add_argument('--file_with_video_paths',
                   help='.txt file where each line is a path')
    p.add_argument('--video_dir', type=str, help='dir of videos')
    p.add_argument('--flow_dir', type=str,
                   help='dir of optical flow: [flow_dir]/[video id]/flow_(x|y)_*.jpg')
    p.add_argument('--device_ids', type=int, nargs='+',
                   help='space-separated device ids')
    p.add_argument('--cpu', action='store_true', help='use cpu only')
    p.add_argument('--tmp_path', default='./tmp')
    p.add_argument('--keep_tmp_files', action='store_true', default=False)
    p.add_argument('--on_extraction', default='print',
                   choices=['print', 'save_numpy', 'save_pickle',
                            'save_jpg'])
    p.add_argument('--output_path', default='./output')
    p.add_argument('--output_direct', action='store_true')
    p.add_argument('--extraction_fps', type=float)
    p.add_argument('--extract_method', type=str,
                   help='"uni_N" (N uniform frames) or "fix_N" (N fps)')
    p.add_argument('--stack_size', type=int)
    p.add_argument('--step_size', type=int)
    p.add_argument('--streams', nargs='+', choices=['flow', 'rgb'])
    p.add_argument('--flow_type', choices=['raft', 'pwc', 'flow'], default='pwc')
    p.add_argument('--batch_size', type=int, default=1)
    p.add_argument('--resize_to_larger_edge', dest='resize_to_smaller_edge',
                   action='store_false', default=True)
    p.add_argument('--side_size', type=int)
    p.add_argument('--show_pred', action='store_true', default=False)
    # ---- additive (not in the reference)
    p.add_argument('--dtype', choices=['auto', 'fp32', 'bf16'], default='auto')
    p.add_argument('--gather_features', action='store_true', default=False)
    p.add_argument('--temporal_parallel', action='store_true', default=False,
                   help='shard each video\'s sliding windows across ranks '
                        '(i3d / r21d_rgb); exact — windows are independent')
    p.add_argument('--resume', action='store_true', default=False,
                   help='skip videos whose outputs already exist')
    p.add_argument('--profile', action='store_true', default=False)
    p.add_argument('--seed', type=int, default=0)
    p.add_argument('--weights_path', type=str, default=None)
    return p


def main(argv: Optional[List[str]] = None) -> None:
    args = build_parser().parse_args(argv)
    cfg = Config.coerce(args)
    if cfg.on_extraction in ('save_numpy', 'save_pickle'):
        print(f'Saving features to {cfg.output_path}')
    if cfg.keep_tmp_files:
        print(f'Keeping temp files in {cfg.tmp_path}')
    sanity_check(cfg)
    run_extraction(cfg)


if __name__ == '__main__':
    main()
