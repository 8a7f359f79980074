"""io.ffmpeg: command construction and naming conventions, with the binary
mocked (no ffmpeg in this environment — reference utils/utils.py:207-276
parity is in the command shapes and tmp naming)."""
import os

import pytest

from video_features_amd.io import ffmpeg as FF


@pytest.fixture
def mock_ffmpeg(monkeypatch, tmp_path):
    calls = []

    def fake_which():
        return '/usr/bin/ffmpeg'

    def fake_run(cmd, check=False, **kw):
        calls.append(cmd)
        # create the output file the command names (always the last arg)
        open(cmd[-1], 'wb').write(b'')
        class R:
            returncode = 0
        return R()

    monkeypatch.setattr(FF, 'which_ffmpeg', fake_which)
    monkeypatch.setattr(FF.subprocess, 'run', fake_run)
    return calls


def test_which_ffmpeg_none_is_graceful(monkeypatch):
    monkeypatch.setattr(FF.shutil, 'which', lambda _: None)
    assert FF.which_ffmpeg() is None
    with pytest.raises(RuntimeError, match='ffmpeg is not installed'):
        FF.require_ffmpeg()


def test_reencode_naming_and_flags(mock_ffmpeg, tmp_path):
    out = FF.reencode_video_with_diff_fps('/videos/clip.mp4',
                                          str(tmp_path), 12.5)
    # reference naming: {stem}_new_fps.mp4 (utils/utils.py:222-244)
    assert out == os.path.join(str(tmp_path), 'clip_new_fps.mp4')
    cmd = mock_ffmpeg[0]
    assert '-filter:v' in cmd and 'fps=fps=12.5' in cmd
    assert '-i' in cmd and '/videos/clip.mp4' in cmd


def test_wav_extraction_two_step(mock_ffmpeg, tmp_path):
    wav, aac = FF.extract_wav_from_video('/videos/clip.mp4', str(tmp_path))
    assert wav.endswith('clip.wav') and aac.endswith('clip.aac')
    # two-step: mp4 -> aac (codec copy) -> wav (reference utils/utils.py:247-276)
    assert len(mock_ffmpeg) == 2
    assert '-acodec' in mock_ffmpeg[0] and 'copy' in mock_ffmpeg[0]
    assert mock_ffmpeg[1][-1] == wav


def test_decode_to_y4m_command(mock_ffmpeg, tmp_path):
    out = FF.decode_to_y4m('/videos/clip.mkv', str(tmp_path), 10.0)
    assert out.endswith('clip.y4m')
    cmd = mock_ffmpeg[0]
    assert 'yuv420p' in cmd
    assert any('fps=10.0' in str(c) for c in cmd)
