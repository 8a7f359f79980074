"""VideoReader — the single decode facade for every extractor.

The reference used three different decoders (mmcv.VideoReader, OpenCV
VideoCapture, torchvision.io.read_video — reference utils/utils.py:310,
extract_resnet.py:121, extract_r21d.py:102).  Here decode is one component
with pluggable backends selected by content:

========================  =============================================
input                     backend
========================  =============================================
``*.y4m``                 native YUV4MPEG2 (io/y4m.py, zero deps)
``*.avi`` (MJPEG)         native RIFF demux + PIL JPEG (io/avi.py)
directory of images       sorted PIL decode (also used for flow JPG dirs)
``*.npy`` / ``*.npz``     (T,H,W,3) uint8 tensor video (+ 'fps' key in npz)
``*.gif``                 PIL
anything else             ffmpeg subprocess → .y4m transcode (needs ffmpeg)
========================  =============================================

All backends expose: ``frame_count``, ``fps``, ``height``, ``width``,
``read_frame(i) -> (H,W,3) uint8 RGB``, ``read_frames(indices)``.
"""
from __future__ import annotations

import os
import re
from pathlib import Path
from typing import Optional

import numpy as np

from .avi import AVIMJPEGReader
from .y4m import Y4MReader
from . import ffmpeg as ffmpeg_mod

IMG_EXTS = ('.jpg', '.jpeg', '.png', '.bmp')


class ImageDirReader:
    """A directory of image frames, sorted naturally (frame_0001.jpg, ...)."""

    def __init__(self, path: str, fps: float = 25.0, pattern: Optional[str] = None):
        self.path = path
        self.fps = fps
        names = [n for n in os.listdir(path) if n.lower().endswith(IMG_EXTS)]
        if pattern:
            rx = re.compile(pattern)
            names = [n for n in names if rx.search(n)]
        if not names:
            raise ValueError(f'{path}: no image frames found')
        names.sort(key=lambda n: [int(t) if t.isdigit() else t
                                  for t in re.split(r'(\d+)', n)])
        self._files = [os.path.join(path, n) for n in names]
        self.frame_count = len(self._files)
        first = self.read_frame(0)
        self.height, self.width = first.shape[:2]

    def read_frame(self, idx: int) -> np.ndarray:
        from PIL import Image
        img = Image.open(self._files[idx]).convert('RGB')
        return np.asarray(img)

    def read_frames(self, indices) -> np.ndarray:
        return np.stack([self.read_frame(int(i)) for i in indices])


class TensorVideoReader:
    """.npy / .npz holding (T, H, W, 3) uint8 RGB (npz may carry 'fps')."""

    def __init__(self, path: str, fps: float = 25.0):
        self.path = path
        if path.endswith('.npz'):
            with np.load(path) as z:
                self._frames = z['frames']
                self.fps = float(z['fps']) if 'fps' in z else fps
        else:
            self._frames = np.load(path)
            self.fps = fps
        if self._frames.ndim != 4 or self._frames.shape[-1] != 3:
            raise ValueError(
                f'{path}: expected (T,H,W,3) array, got {self._frames.shape}')
        self.frame_count = self._frames.shape[0]
        self.height, self.width = self._frames.shape[1:3]

    def read_frame(self, idx: int) -> np.ndarray:
        return np.ascontiguousarray(self._frames[idx])

    def read_frames(self, indices) -> np.ndarray:
        return np.ascontiguousarray(self._frames[np.asarray(indices, np.int64)])


class GIFReader:
    def __init__(self, path: str):
        from PIL import Image, ImageSequence
        self.path = path
        img = Image.open(path)
        self._frames = [np.asarray(f.convert('RGB'))
                        for f in ImageSequence.Iterator(img)]
        dur = img.info.get('duration', 40) or 40
        self.fps = 1000.0 / dur
        self.frame_count = len(self._frames)
        self.height, self.width = self._frames[0].shape[:2]

    def read_frame(self, idx: int) -> np.ndarray:
        return self._frames[idx]

    def read_frames(self, indices) -> np.ndarray:
        return np.stack([self._frames[int(i)] for i in indices])


def open_video(path: str, tmp_path: str = './tmp',
               extraction_fps: Optional[float] = None):
    """Open any supported video.  ``extraction_fps`` re-times the stream: for
    the ffmpeg backend via transcode, for native backends via index
    resampling (wrapping the reader in :class:`ResampledReader`)."""
    p = str(path)
    if os.path.isdir(p):
        reader = ImageDirReader(p)
    else:
        ext = Path(p).suffix.lower()
        if ext == '.y4m':
            reader = Y4MReader(p)
        elif ext == '.avi':
            reader = AVIMJPEGReader(p)
        elif ext in ('.npy', '.npz'):
            reader = TensorVideoReader(p)
        elif ext == '.gif':
            reader = GIFReader(p)
        elif ext in IMG_EXTS:
            reader = TensorVideoReader.__new__(TensorVideoReader)  # pragma: no cover
            raise ValueError(f'{p}: single images are not videos')
        else:
            # compressed container — bridge through ffmpeg if present
            y4m = ffmpeg_mod.decode_to_y4m(p, tmp_path, extraction_fps)
            return Y4MReader(y4m)
    if extraction_fps and abs(extraction_fps - reader.fps) > 1e-6:
        reader = ResampledReader(reader, extraction_fps)
    # lazily fill height/width for readers that don't probe eagerly
    if not hasattr(reader, 'height'):
        f0 = reader.read_frame(0)
        reader.height, reader.width = f0.shape[:2]
    return reader


class ResampledReader:
    """Re-times a native reader to a target fps by nearest-index sampling —
    the in-memory analog of the reference's ffmpeg re-encode
    (reference utils/utils.py:222-244)."""

    def __init__(self, inner, target_fps: float):
        self.inner = inner
        self.fps = float(target_fps)
        duration = inner.frame_count / inner.fps
        self.frame_count = max(1, int(round(duration * target_fps)))
        self.height = getattr(inner, 'height', None)
        self.width = getattr(inner, 'width', None)
        scale = inner.fps / target_fps
        self._map = np.clip((np.arange(self.frame_count) * scale).round(),
                            0, inner.frame_count - 1).astype(np.int64)
        if self.height is None:
            f0 = inner.read_frame(0)
            self.height, self.width = f0.shape[:2]

    def read_frame(self, idx: int) -> np.ndarray:
        return self.inner.read_frame(int(self._map[idx]))

    def read_frames(self, indices) -> np.ndarray:
        return self.inner.read_frames(self._map[np.asarray(indices, np.int64)])
