"""Minimal YUV4MPEG2 (.y4m) reader/writer — a zero-dependency raw video format.

Lets the framework own a decode path with no ffmpeg/mmcv/OpenCV in the image
(the reference shells out to ffmpeg and uses mmcv.VideoReader — reference
utils/utils.py:207-276, 310).  Y4M is uncompressed 4:2:0 YUV with a text
header, trivially parseable, and ffmpeg can produce it anywhere, so it doubles
as the interchange format between an external ffmpeg and this framework.
"""
from __future__ import annotations

import os
from typing import Tuple

import numpy as np

_MAGIC = b'YUV4MPEG2'


def _yuv_to_rgb(y: np.ndarray, u_full: np.ndarray,
                v_full: np.ndarray) -> np.ndarray:
    """BT.601 full-swing conversion (JPEG/JFIF convention), full-res
    chroma planes."""
    yf = y.astype(np.float32)
    uf = u_full.astype(np.float32) - 128.0
    vf = v_full.astype(np.float32) - 128.0
    r = yf + 1.402 * vf
    g = yf - 0.344136 * uf - 0.714136 * vf
    b = yf + 1.772 * uf
    return np.clip(np.stack([r, g, b], axis=-1), 0, 255).astype(np.uint8)


def _yuv420_to_rgb(y: np.ndarray, u: np.ndarray, v: np.ndarray) -> np.ndarray:
    h, w = y.shape
    u_full = np.repeat(np.repeat(u, 2, axis=0), 2, axis=1)[:h, :w]
    v_full = np.repeat(np.repeat(v, 2, axis=0), 2, axis=1)[:h, :w]
    return _yuv_to_rgb(y, u_full, v_full)


def _rgb_to_yuv(rgb: np.ndarray):
    f = rgb.astype(np.float32)
    r, g, b = f[..., 0], f[..., 1], f[..., 2]
    y = 0.299 * r + 0.587 * g + 0.114 * b
    u = -0.168736 * r - 0.331264 * g + 0.5 * b + 128.0
    v = 0.5 * r - 0.418688 * g - 0.081312 * b + 128.0
    return (np.clip(y, 0, 255).astype(np.uint8),
            np.clip(u, 0, 255).astype(np.uint8),
            np.clip(v, 0, 255).astype(np.uint8))


def _rgb_to_yuv420(rgb: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    f = rgb.astype(np.float32)
    r, g, b = f[..., 0], f[..., 1], f[..., 2]
    y = 0.299 * r + 0.587 * g + 0.114 * b
    u = -0.168736 * r - 0.331264 * g + 0.5 * b + 128.0
    v = 0.5 * r - 0.418688 * g - 0.081312 * b + 128.0
    y8 = np.clip(y, 0, 255).astype(np.uint8)
    # average 2x2 blocks for chroma subsampling (pad odd dims)
    h, w = y8.shape
    hp, wp = h + (h & 1), w + (w & 1)
    up = np.zeros((hp, wp), np.float32)
    vp = np.zeros((hp, wp), np.float32)
    up[:h, :w], vp[:h, :w] = u, v
    if h & 1:
        up[h, :w], vp[h, :w] = u[h - 1], v[h - 1]
    if w & 1:
        up[:h, w], vp[:h, w] = u[:, w - 1], v[:, w - 1]
    u4 = up.reshape(hp // 2, 2, wp // 2, 2).mean(axis=(1, 3))
    v4 = vp.reshape(hp // 2, 2, wp // 2, 2).mean(axis=(1, 3))
    return y8, np.clip(u4, 0, 255).astype(np.uint8), np.clip(v4, 0, 255).astype(np.uint8)


class Y4MReader:
    """Index-addressable reader over a .y4m file (frames memory-mapped lazily)."""

    def __init__(self, path: str):
        self.path = path
        with open(path, 'rb') as f:
            header = f.readline()
        if not header.startswith(_MAGIC):
            raise ValueError(f'{path}: not a YUV4MPEG2 file')
        self.width = self.height = 0
        fps_num, fps_den = 25, 1
        colorspace = 'C420'
        for tok in header.split()[1:]:
            t = tok.decode('ascii', 'replace')
            if t.startswith('W'):
                self.width = int(t[1:])
            elif t.startswith('H'):
                self.height = int(t[1:])
            elif t.startswith('F'):
                num, den = t[1:].split(':')
                fps_num, fps_den = int(num), int(den)
            elif t.startswith('C'):
                colorspace = t
        if colorspace.startswith('C420'):
            self._c444 = False
        elif colorspace.startswith('C444'):
            self._c444 = True
        else:
            raise ValueError(
                f'{path}: only 4:2:0 / 4:4:4 y4m supported, got {colorspace}')
        self.fps = fps_num / fps_den
        self._header_len = len(header)
        y_sz = self.width * self.height
        c_sz = y_sz if self._c444 else \
            ((self.width + 1) // 2) * ((self.height + 1) // 2)
        self._frame_data = y_sz + 2 * c_sz
        self._y_sz, self._c_sz = y_sz, c_sz
        # frame record = b'FRAME...\n' + planes; assume constant FRAME header len
        with open(path, 'rb') as f:
            f.seek(self._header_len)
            fh = f.readline()
        if not fh.startswith(b'FRAME'):
            raise ValueError(f'{path}: malformed y4m (no FRAME marker)')
        self._frame_hdr_len = len(fh)
        total = os.path.getsize(path) - self._header_len
        rec = self._frame_hdr_len + self._frame_data
        self.frame_count = total // rec
        self._rec = rec

    def read_frame(self, idx: int) -> np.ndarray:
        if not (0 <= idx < self.frame_count):
            raise IndexError(f'frame {idx} out of range [0, {self.frame_count})')
        off = self._header_len + idx * self._rec + self._frame_hdr_len
        with open(self.path, 'rb') as f:
            f.seek(off)
            raw = f.read(self._frame_data)
        w, h = self.width, self.height
        cw, ch = (w, h) if self._c444 else ((w + 1) // 2, (h + 1) // 2)
        y = np.frombuffer(raw, np.uint8, self._y_sz).reshape(h, w)
        u = np.frombuffer(raw, np.uint8, self._c_sz, offset=self._y_sz).reshape(ch, cw)
        v = np.frombuffer(raw, np.uint8, self._c_sz, offset=self._y_sz + self._c_sz).reshape(ch, cw)
        if self._c444:
            return _yuv_to_rgb(y, u, v)
        return _yuv420_to_rgb(y, u, v)

    def read_frames(self, indices) -> np.ndarray:
        """Batched read + ONE vectorized YUV→RGB conversion over all
        requested frames (the per-frame numpy conversion dominated the
        file-decode path: 470 → measured ~2× on the ResNet extractor)."""
        idxs = [int(i) for i in indices]
        for i in idxs:
            if not (0 <= i < self.frame_count):
                raise IndexError(
                    f'frame {i} out of range [0, {self.frame_count})')
        n = len(idxs)
        raws = np.empty((n, self._frame_data), np.uint8)
        with open(self.path, 'rb') as f:
            for j, idx in enumerate(idxs):
                f.seek(self._header_len + idx * self._rec +
                       self._frame_hdr_len)
                f.readinto(memoryview(raws[j]))
        h, w = self.height, self.width
        cw, ch = (w, h) if self._c444 else ((w + 1) // 2, (h + 1) // 2)
        y = raws[:, :self._y_sz].reshape(n, h, w)
        u = raws[:, self._y_sz:self._y_sz + self._c_sz].reshape(n, ch, cw)
        v = raws[:, self._y_sz + self._c_sz:].reshape(n, ch, cw)
        if not self._c444:
            u = np.repeat(np.repeat(u, 2, axis=1), 2, axis=2)[:, :h, :w]
            v = np.repeat(np.repeat(v, 2, axis=1), 2, axis=2)[:, :h, :w]
        return _yuv_to_rgb(y, u, v)


def write_y4m(path: str, frames: np.ndarray, fps: float = 25.0,
              colorspace: str = 'C420jpeg') -> None:
    """Write (T, H, W, 3) uint8 RGB frames as y4m.  ``C420jpeg`` matches
    typical video chroma (lossy roundtrip on sharp chroma edges);
    ``C444`` keeps full-resolution chroma."""
    frames = np.asarray(frames)
    if frames.ndim != 4 or frames.shape[-1] != 3:
        raise ValueError(f'expected (T,H,W,3), got {frames.shape}')
    c444 = colorspace.startswith('C444')
    t, h, w, _ = frames.shape
    from fractions import Fraction
    fr = Fraction(fps).limit_denominator(1001)
    with open(path, 'wb') as f:
        f.write(f'YUV4MPEG2 W{w} H{h} F{fr.numerator}:{fr.denominator} '
                f'Ip A1:1 {colorspace}\n'.encode('ascii'))
        for i in range(t):
            if c444:
                y, u, v = _rgb_to_yuv(frames[i])
            else:
                y, u, v = _rgb_to_yuv420(frames[i])
            f.write(b'FRAME\n')
            f.write(y.tobytes())
            f.write(u.tobytes())
            f.write(v.tobytes())
