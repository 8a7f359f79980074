"""ffmpeg subprocess helpers (used only when an ffmpeg binary exists).

The reference shells out to ffmpeg for fps re-encoding and audio extraction
(reference utils/utils.py:207-276).  This module keeps that capability with
the same tmp-file naming conventions, but the rest of the framework never
*requires* ffmpeg: native formats (.y4m, MJPEG .avi, frame dirs, .npy) decode
with zero external binaries (see io/video.py).
"""
from __future__ import annotations

import os
import shutil
import subprocess
from pathlib import Path
from typing import Optional, Tuple


def which_ffmpeg() -> Optional[str]:
    """Absolute path of ffmpeg, or None (reference utils/utils.py:207-219
    raises; we return None so callers can fall back to native decode)."""
    return shutil.which('ffmpeg')


def require_ffmpeg() -> str:
    path = which_ffmpeg()
    if path is None:
        raise RuntimeError(
            'ffmpeg is not installed on this host. Compressed formats (.mp4, '
            '.mkv, ...) need it; native formats (.y4m, MJPEG .avi, frame '
            'directories, .npy/.npz) do not.')
    return path


def reencode_video_with_diff_fps(video_path: str, tmp_path: str,
                                 extraction_fps: float) -> str:
    """Re-encode a video at a new constant fps into ``tmp_path``; returns the
    new file path (reference utils/utils.py:222-244 semantics/naming)."""
    ffmpeg = require_ffmpeg()
    os.makedirs(tmp_path, exist_ok=True)
    stem = Path(video_path).stem
    out = os.path.join(tmp_path, f'{stem}_new_fps.mp4')
    cmd = [ffmpeg, '-hide_banner', '-loglevel', 'panic', '-y',
           '-i', video_path, '-filter:v', f'fps=fps={extraction_fps}', out]
    subprocess.run(cmd, check=True)
    return out


def extract_wav_from_video(video_path: str, tmp_path: str) -> Tuple[str, str]:
    """Extract audio to a 16-bit PCM wav via an intermediate .aac, mirroring
    the reference's two-step pipeline (reference utils/utils.py:247-276).
    Returns (wav_path, aac_path)."""
    ffmpeg = require_ffmpeg()
    os.makedirs(tmp_path, exist_ok=True)
    stem = Path(video_path).stem
    aac = os.path.join(tmp_path, f'{stem}.aac')
    wav = os.path.join(tmp_path, f'{stem}.wav')
    subprocess.run([ffmpeg, '-hide_banner', '-loglevel', 'panic', '-y',
                    '-i', video_path, '-acodec', 'copy', aac], check=True)
    subprocess.run([ffmpeg, '-hide_banner', '-loglevel', 'panic', '-y',
                    '-i', aac, wav], check=True)
    return wav, aac


def decode_to_y4m(video_path: str, tmp_path: str,
                  extraction_fps: Optional[float] = None) -> str:
    """Transcode any ffmpeg-readable video into the framework's native .y4m
    (the bridge between external codecs and the zero-dependency readers)."""
    ffmpeg = require_ffmpeg()
    os.makedirs(tmp_path, exist_ok=True)
    stem = Path(video_path).stem
    out = os.path.join(tmp_path, f'{stem}.y4m')
    cmd = [ffmpeg, '-hide_banner', '-loglevel', 'panic', '-y', '-i', video_path]
    if extraction_fps:
        cmd += ['-filter:v', f'fps=fps={extraction_fps}']
    cmd += ['-pix_fmt', 'yuv420p', out]
    subprocess.run(cmd, check=True)
    return out
