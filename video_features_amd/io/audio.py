"""Audio IO: stdlib WAV decode + polyphase resampling (no soundfile/resampy).

The reference reads wavs via soundfile and resamples via resampy
(reference models/vggish_torch/vggish_src/vggish_input.py:84-98); both are
absent from this image, so the framework owns the path: ``wave`` (stdlib) +
numpy for decode, scipy.signal.resample_poly for resampling.
"""
from __future__ import annotations

import wave
from pathlib import Path
from typing import Tuple

import numpy as np

from . import ffmpeg as ffmpeg_mod


def read_wav(path: str) -> Tuple[np.ndarray, int]:
    """Read a PCM wav file → (float32 samples in [-1, 1] shaped (T,) mono or
    (T, C), sample_rate)."""
    with wave.open(path, 'rb') as w:
        sr = w.getframerate()
        n = w.getnframes()
        ch = w.getnchannels()
        sw = w.getsampwidth()
        raw = w.readframes(n)
    if sw == 2:
        data = np.frombuffer(raw, np.int16).astype(np.float32) / 32768.0
    elif sw == 1:
        data = (np.frombuffer(raw, np.uint8).astype(np.float32) - 128.0) / 128.0
    elif sw == 4:
        data = np.frombuffer(raw, np.int32).astype(np.float32) / 2147483648.0
    else:
        raise ValueError(f'{path}: unsupported sample width {sw}')
    if ch > 1:
        data = data.reshape(-1, ch)
    return data, sr


def write_wav(path: str, samples: np.ndarray, sr: int) -> None:
    """Write float32 [-1,1] samples as 16-bit PCM."""
    samples = np.asarray(samples)
    if samples.ndim == 1:
        ch = 1
    else:
        ch = samples.shape[1]
    pcm = np.clip(samples * 32767.0, -32768, 32767).astype(np.int16)
    with wave.open(path, 'wb') as w:
        w.setnchannels(ch)
        w.setsampwidth(2)
        w.setframerate(sr)
        w.writeframes(pcm.tobytes())


def to_mono(samples: np.ndarray) -> np.ndarray:
    return samples.mean(axis=1) if samples.ndim == 2 else samples


def resample(samples: np.ndarray, sr: int, target_sr: int) -> np.ndarray:
    if sr == target_sr:
        return samples
    from math import gcd
    from scipy.signal import resample_poly
    g = gcd(int(sr), int(target_sr))
    return resample_poly(samples, target_sr // g, sr // g).astype(np.float32)


def load_audio_for_video(video_path: str, tmp_path: str = './tmp',
                         keep_tmp_files: bool = False) -> Tuple[np.ndarray, int, list]:
    """Get mono float32 audio for a video input.  Resolution order:

    1. the input itself is a ``.wav``;
    2. a sidecar ``<stem>.wav`` next to the video (the native, ffmpeg-free
       route for authored fixtures / pre-demuxed audio);
    3. ffmpeg extraction (mp4 → aac → wav, the reference pipeline —
       reference utils/utils.py:247-276).

    Returns (samples, sample_rate, tmp_files_created).
    """
    p = Path(video_path)
    tmp_files = []
    if p.suffix.lower() == '.wav':
        wav_path = str(p)
    else:
        sidecar = p.with_suffix('.wav')
        if sidecar.exists():
            wav_path = str(sidecar)
        else:
            wav_path, aac_path = ffmpeg_mod.extract_wav_from_video(str(p), tmp_path)
            tmp_files = [wav_path, aac_path]
    samples, sr = read_wav(wav_path)
    return to_mono(samples), sr, tmp_files
