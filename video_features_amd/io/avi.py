"""Minimal RIFF/AVI MJPEG demuxer + muxer (pure Python; JPEG via PIL).

Gives the framework a *compressed* native video format that round-trips with
no external binaries: frames are ordinary JPEGs inside an AVI container.  Any
ffmpeg can produce such files (``-c:v mjpeg``), and the demuxer here needs
only stdlib + PIL.  This replaces the reference's hard dependency on
mmcv/OpenCV decoders (reference utils/utils.py:310, extract_resnet.py:121).
"""
from __future__ import annotations

import io
import struct
from typing import List

import numpy as np


def _fourcc(tag: bytes) -> bytes:
    assert len(tag) == 4
    return tag


class AVIMJPEGReader:
    """Index-addressable MJPEG-in-AVI reader.

    Scans the 'movi' list once at open for '..dc'/'..db' chunk offsets; frames
    decode lazily through PIL.
    """

    def __init__(self, path: str):
        self.path = path
        self.fps = 25.0
        self.width = self.height = 0
        self._offsets: List[tuple] = []   # (file_offset, size)
        with open(path, 'rb') as f:
            riff, size, ftype = struct.unpack('<4sI4s', f.read(12))
            if riff != b'RIFF' or ftype != b'AVI ':
                raise ValueError(f'{path}: not an AVI file')
            self._scan(f, 12, 8 + size)
        self.frame_count = len(self._offsets)

    def _scan(self, f, pos: int, end: int) -> None:
        while pos + 8 <= end:
            f.seek(pos)
            hdr = f.read(8)
            if len(hdr) < 8:
                break
            tag, sz = struct.unpack('<4sI', hdr)
            if tag == b'LIST':
                ltype = f.read(4)
                if ltype in (b'hdrl', b'movi', b'strl'):
                    self._scan(f, pos + 12, pos + 8 + sz)
            elif tag == b'avih':
                data = f.read(sz)
                usec_per_frame = struct.unpack('<I', data[0:4])[0]
                if usec_per_frame:
                    self.fps = 1e6 / usec_per_frame
                self.width = struct.unpack('<I', data[32:36])[0]
                self.height = struct.unpack('<I', data[36:40])[0]
            elif tag[2:4] in (b'dc', b'db'):
                self._offsets.append((pos + 8, sz))
            pos += 8 + sz + (sz & 1)   # chunks are word-aligned

    def read_frame(self, idx: int) -> np.ndarray:
        from PIL import Image
        if not (0 <= idx < self.frame_count):
            raise IndexError(f'frame {idx} out of range [0, {self.frame_count})')
        off, sz = self._offsets[idx]
        with open(self.path, 'rb') as f:
            f.seek(off)
            data = f.read(sz)
        img = Image.open(io.BytesIO(data)).convert('RGB')
        return np.asarray(img)

    def read_frames(self, indices) -> np.ndarray:
        return np.stack([self.read_frame(int(i)) for i in indices])


def write_avi_mjpeg(path: str, frames: np.ndarray, fps: float = 25.0,
                    quality: int = 90) -> None:
    """Write (T, H, W, 3) uint8 RGB frames as an MJPEG AVI file."""
    from PIL import Image
    frames = np.asarray(frames)
    t, h, w, _ = frames.shape
    jpegs = []
    for i in range(t):
        buf = io.BytesIO()
        Image.fromarray(frames[i]).save(buf, format='JPEG', quality=quality)
        jpegs.append(buf.getvalue())

    def chunk(tag: bytes, data: bytes) -> bytes:
        pad = b'\x00' if len(data) & 1 else b''
        return tag + struct.pack('<I', len(data)) + data + pad

    def lst(ltype: bytes, data: bytes) -> bytes:
        return chunk(b'LIST', ltype + data)

    usec = int(round(1e6 / fps))
    max_jpeg = max(len(j) for j in jpegs)
    # MainAVIHeader: usec/frame, maxbytes/sec, pad, flags(HASINDEX), frames,
    # initial, streams, bufsize, W, H, reserved[4]
    avih = struct.pack('<10I4I', usec, max_jpeg * int(fps + 1), 0, 0x10, t, 0, 1,
                       max_jpeg, w, h, 0, 0, 0, 0)
    # AVIStreamHeader: fccType, fccHandler, flags, prio+lang, initial, scale,
    # rate, start, length, bufsize, quality, samplesize, rcFrame
    strh = (b'vids' + b'MJPG' + struct.pack('<IIIIIIIIII', 0, 0, 0, 1000,
            int(round(fps * 1000)), 0, t, max_jpeg, 0, 0)
            + struct.pack('<4h', 0, 0, w, h))
    # BITMAPINFOHEADER
    strf = struct.pack('<IiiHH4sIiiII', 40, w, h, 1, 24, b'MJPG',
                       w * h * 3, 0, 0, 0, 0)
    hdrl = lst(b'hdrl', chunk(b'avih', avih)
               + lst(b'strl', chunk(b'strh', strh) + chunk(b'strf', strf)))
    movi_chunks = b''.join(chunk(b'00dc', j) for j in jpegs)
    movi = lst(b'movi', movi_chunks)
    # idx1 (offsets relative to 'movi' fourcc position)
    idx_entries, off = [], 4
    for j in jpegs:
        idx_entries.append(struct.pack('<4sIII', b'00dc', 0x10, off, len(j)))
        off += 8 + len(j) + (len(j) & 1)
    idx1 = chunk(b'idx1', b''.join(idx_entries))
    body = b'AVI ' + hdrl + movi + idx1
    with open(path, 'wb') as f:
        f.write(b'RIFF' + struct.pack('<I', len(body)) + body)
