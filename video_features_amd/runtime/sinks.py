"""Output sinks: print / save_numpy / save_pickle (+ flow-image save).

Reproduces the reference's naming and behaviour
(reference utils/utils.py:50-114):

- meta keys ``fps`` and ``timestamps_ms`` ride along inside the feats dict
  but are saved under their own names;
- file naming: ``{stem}_{key}.{ext}``, or ``{stem}.{ext}`` when
  ``output_direct`` and the key equals the feature type;
- 'print' shows shape + max/mean/min per tensor.

The reference's ``save_jpg`` flow branch is dead code with an
iterate-over-int bug (reference utils/utils.py:96-112); here flow images are
saved correctly as .npy through the normal path.
"""
from __future__ import annotations

import os
import pickle
from pathlib import Path
from typing import Dict

import numpy as np

META_KEYS = ('fps', 'timestamps_ms')


def make_output_path(output_path: str, feature_type: str, output_direct: bool) -> str:
    """Reference layout: features land in ``{output_path}/{feature_type}/``
    unless ``output_direct`` (reference extract_clip.py:31-35)."""
    return output_path if output_direct else os.path.join(output_path, feature_type)


def output_file(out_dir: str, video_path: str, key: str, feature_type: str,
                output_direct: bool, ext: str) -> str:
    stem = Path(video_path).stem
    if output_direct and key == feature_type:
        name = f'{stem}.{ext}'
    else:
        name = f'{stem}_{key.replace("/", "_")}.{ext}'
    return os.path.join(out_dir, name)


def action_on_extraction(feats_dict: Dict[str, np.ndarray], video_path: str,
                         output_path: str, on_extraction: str = 'print',
                         output_direct: bool = False,
                         feature_type: str = '') -> None:
    if on_extraction == 'print':
        print(f'{video_path}:')
        for key, value in feats_dict.items():
            arr = np.asarray(value)
            if arr.dtype == object or arr.ndim == 0:
                print(f'  {key}: {value}')
            else:
                print(f'  {key}: shape {arr.shape}, '
                      f'max {arr.max():.5f}, mean {arr.mean():.5f}, min {arr.min():.5f}')
        return

    os.makedirs(output_path, exist_ok=True)
    if on_extraction == 'save_numpy':
        for key, value in feats_dict.items():
            if key in META_KEYS:
                continue
            fpath = output_file(output_path, video_path, key, feature_type,
                                output_direct, 'npy')
            np.save(fpath, np.asarray(value))
    elif on_extraction == 'save_pickle':
        fpath = output_file(output_path, video_path, feature_type or 'feats',
                            feature_type, output_direct, 'pkl')
        with open(fpath, 'wb') as f:
            pickle.dump(feats_dict, f)
    elif on_extraction == 'save_jpg':
        # flow-image export: (T, 2, H, W) flow fields become per-frame
        # Middlebury visualizations (the reference documents this sink but
        # ships a dead branch with an iterate-over-int bug, reference
        # utils/utils.py:96-112; here it works)
        from ..utils.flow_viz import flow_to_image
        from PIL import Image
        stem = Path(video_path).stem
        for key, value in feats_dict.items():
            if key in META_KEYS:
                continue
            arr = np.asarray(value)
            if arr.ndim != 4 or arr.shape[1] != 2:
                raise ValueError(
                    f'save_jpg expects (T, 2, H, W) flow features; '
                    f'{key} has shape {arr.shape} — use save_numpy')
            vdir = os.path.join(output_path, f'{stem}_{key}')
            os.makedirs(vdir, exist_ok=True)
            for t in range(arr.shape[0]):
                img = flow_to_image(arr[t].transpose(1, 2, 0))
                Image.fromarray(img).save(
                    os.path.join(vdir, f'flow_{t:06d}.jpg'))
    else:
        raise ValueError(f'unknown on_extraction {on_extraction!r}')


def outputs_exist(feats_keys, video_path: str, output_path: str,
                  on_extraction: str, output_direct: bool,
                  feature_type: str) -> bool:
    """Resume support (a capability the reference lacks — its outputs silently
    overwrite): True when every expected output file already exists."""
    if on_extraction == 'print':
        return False
    if on_extraction == 'save_pickle':
        return os.path.exists(output_file(output_path, video_path,
                                          feature_type or 'feats', feature_type,
                                          output_direct, 'pkl'))
    files = [output_file(output_path, video_path, k, feature_type,
                         output_direct, 'npy')
             for k in feats_keys if k not in META_KEYS]
    return bool(files) and all(os.path.exists(f) for f in files)
