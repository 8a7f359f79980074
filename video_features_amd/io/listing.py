"""Input listing — resolve user-provided sources into a concrete path list.

Reproduces the reference's precedence and pairing behaviour
(reference utils/utils.py:153-204):

- ``file_with_video_paths`` (txt, one path per line, blanks skipped), else
- ``video_dir`` (every file inside), else
- ``video_paths`` (explicit list);
- optional pairing with pre-computed flow: ``flow_paths`` index-aligned, or
  ``flow_dir`` joined by video stem → ``(video, flow_dir_for_video)`` tuples;
- existence check up-front (fail before any GPU work).
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import List, Tuple, Union

from ..config import Config

PathOrPair = Union[str, Tuple[str, str]]

VIDEO_EXTS = ('.mp4', '.mkv', '.webm', '.mov', '.avi', '.y4m', '.gif',
              '.npy', '.npz', '.wav')


def form_list_from_user_input(cfg) -> List[PathOrPair]:
    cfg = Config.coerce(cfg)
    if cfg.file_with_video_paths:
        with open(cfg.file_with_video_paths) as f:
            paths = [ln.strip() for ln in f.readlines()]
        paths = [p for p in paths if p]
    elif cfg.video_dir:
        names = sorted(os.listdir(cfg.video_dir))
        paths = [os.path.join(cfg.video_dir, n) for n in names]
        paths = [p for p in paths
                 if os.path.isdir(p) or p.lower().endswith(VIDEO_EXTS)]
    elif cfg.video_paths:
        paths = list(cfg.video_paths)
    else:
        raise ValueError('no inputs: provide --video_paths, --video_dir or '
                         '--file_with_video_paths')

    missing = [p for p in paths if not os.path.exists(p)]
    if missing:
        raise FileNotFoundError(f'inputs do not exist: {missing}')

    # pre-computed flow pairing (used by i3d with --flow_type flow)
    if cfg.flow_paths:
        if len(cfg.flow_paths) != len(paths):
            raise ValueError(
                f'flow_paths ({len(cfg.flow_paths)}) must align with videos ({len(paths)})')
        return list(zip(paths, cfg.flow_paths))
    if cfg.flow_dir:
        pairs = []
        for p in paths:
            stem = Path(p).stem
            fdir = os.path.join(cfg.flow_dir, stem)
            if not os.path.isdir(fdir):
                raise FileNotFoundError(f'flow dir missing for {p}: {fdir}')
            pairs.append((p, fdir))
        return pairs
    return paths
