import os

import numpy as np
import pytest

from video_features_amd.io.avi import AVIMJPEGReader, write_avi_mjpeg
from video_features_amd.io.audio import read_wav, to_mono, write_wav, resample
from video_features_amd.io.video import open_video
from video_features_amd.io.y4m import Y4MReader, write_y4m
from tests.conftest import synthetic_frames


def test_y4m_roundtrip(tmp_path, frames16):
    p = str(tmp_path / 'v.y4m')
    write_y4m(p, frames16, fps=30.0)
    r = Y4MReader(p)
    assert r.frame_count == 16
    assert abs(r.fps - 30.0) < 1e-6
    assert (r.height, r.width) == (64, 96)
    out = r.read_frame(3)
    # 4:2:0 chroma subsampling loses a little; luma-dominant error stays small
    err = np.abs(out.astype(int) - frames16[3].astype(int)).mean()
    assert err < 6.0, err


def test_avi_mjpeg_roundtrip(tmp_path, frames16):
    p = str(tmp_path / 'v.avi')
    write_avi_mjpeg(p, frames16, fps=12.0, quality=95)
    r = AVIMJPEGReader(p)
    assert r.frame_count == 16
    assert abs(r.fps - 12.0) < 0.1
    out = r.read_frames([0, 5, 15])
    assert out.shape == (3, 64, 96, 3)
    err = np.abs(out[1].astype(int) - frames16[5].astype(int)).mean()
    assert err < 8.0, err


def test_npz_reader(npz_video, frames16):
    r = open_video(npz_video)
    assert r.frame_count == 16
    assert r.fps == 25.0
    np.testing.assert_array_equal(r.read_frame(7), frames16[7])


def test_image_dir_reader(tmp_path, frames16):
    from PIL import Image
    d = tmp_path / 'frames'
    d.mkdir()
    for i in range(8):
        Image.fromarray(frames16[i]).save(str(d / f'frame_{i:04d}.png'))
    r = open_video(str(d))
    assert r.frame_count == 8
    np.testing.assert_array_equal(r.read_frame(2), frames16[2])


def test_resampled_reader(y4m_video):
    # 16 frames @ 25fps = 0.64 s; re-time to 50 fps → 32 frames
    r = open_video(y4m_video, extraction_fps=50.0)
    assert r.frame_count == 32
    assert r.fps == 50.0
    assert r.read_frame(31).shape == (64, 96, 3)


def test_unsupported_codec_without_ffmpeg(tmp_path):
    from video_features_amd.io import which_ffmpeg
    p = tmp_path / 'fake.mp4'
    p.write_bytes(b'\x00' * 64)
    if which_ffmpeg() is None:
        with pytest.raises(RuntimeError, match='ffmpeg'):
            open_video(str(p))


def test_wav_roundtrip(tmp_path):
    sr = 16000
    t = np.arange(sr) / sr
    sig = (0.5 * np.sin(2 * np.pi * 440 * t)).astype(np.float32)
    p = str(tmp_path / 'a.wav')
    write_wav(p, sig, sr)
    out, sr2 = read_wav(p)
    assert sr2 == sr
    assert np.abs(out - sig).max() < 1e-3


def test_resample():
    sr = 44100
    t = np.arange(sr) / sr
    sig = np.sin(2 * np.pi * 440 * t).astype(np.float32)
    out = resample(sig, sr, 16000)
    assert abs(len(out) - 16000) <= 2


def test_listing(tmp_path, frames16):
    from video_features_amd.config import Config
    from video_features_amd.io.listing import form_list_from_user_input
    v1 = tmp_path / 'a.y4m'
    v2 = tmp_path / 'b.y4m'
    write_y4m(str(v1), frames16, 25)
    write_y4m(str(v2), frames16, 25)
    # video_dir
    cfg = Config(video_dir=str(tmp_path))
    assert form_list_from_user_input(cfg) == [str(v1), str(v2)]
    # file_with_video_paths
    lst = tmp_path / 'list.txt'
    lst.write_text(f'{v1}\n\n{v2}\n')
    cfg = Config(file_with_video_paths=str(lst))
    assert form_list_from_user_input(cfg) == [str(v1), str(v2)]
    # missing file raises up-front
    cfg = Config(video_paths=[str(v1), str(tmp_path / 'nope.y4m')])
    with pytest.raises(FileNotFoundError):
        form_list_from_user_input(cfg)


def test_y4m_c444_roundtrip(tmp_path):
    """C444 y4m keeps full-resolution chroma: random-noise frames roundtrip
    within YUV<->RGB rounding (C420 subsampling would lose ~±240 on sharp
    chroma edges)."""
    from video_features_amd.io.y4m import write_y4m
    rng = np.random.default_rng(0)
    frames = rng.integers(0, 256, (3, 33, 47, 3), dtype=np.uint8)
    p = str(tmp_path / 'c444.y4m')
    write_y4m(p, frames, fps=30.0, colorspace='C444')
    r = open_video(p)
    back = r.read_frames([0, 1, 2])
    err = np.abs(back.astype(int) - frames.astype(int)).max()
    assert err <= 3, err
