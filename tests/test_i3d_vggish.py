import numpy as np
import pytest
import torch

from video_features_amd.config import Config


def test_i3d_tf_same_padding():
    from video_features_amd.models.i3d import _same_pad_1d, tf_same_pad_3d
    # TF-SAME: k=7, s=2, n=224 → total pad 5, split (2, 3) — asymmetric
    assert _same_pad_1d(224, 7, 2) == (2, 3)
    assert _same_pad_1d(64, 3, 1) == (1, 1)
    assert _same_pad_1d(7, 2, 2) == (0, 1)
    x = torch.zeros(1, 1, 8, 10, 12)
    y = tf_same_pad_3d(x, (3, 3, 3), (1, 1, 1))
    assert y.shape == (1, 1, 10, 12, 14)


def test_i3d_shapes_both_streams():
    from video_features_amd.models.i3d import I3D
    torch.manual_seed(0)
    for modality, in_ch in [('rgb', 3), ('flow', 2)]:
        m = I3D(modality=modality).eval()
        with torch.no_grad():
            # small spatial size for CPU test speed; 16-frame clip
            f = m.forward_features(torch.randn(1, in_ch, 16, 112, 112))
            logits = m(torch.randn(1, in_ch, 16, 112, 112))
        assert f.shape == (1, 1024), modality
        assert logits.shape == (1, 400)
        assert torch.isfinite(f).all()


def test_i3d_param_count():
    from video_features_amd.models.i3d import I3D
    n = sum(p.numel() for p in I3D(modality='rgb').parameters())
    # published I3D RGB (400 classes): ~12.3M parameters
    assert 11.5e6 < n < 13.5e6, n


def test_i3d_extractor_end_to_end(tmp_path):
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.i3d import ExtractI3D
    from video_features_amd.io.y4m import write_y4m
    # tiny stacks for CPU: stack 10, step 10, 21-frame video → 2 stacks
    frames = synthetic_frames(t=21, h=64, w=64)
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, frames, fps=25.0)
    cfg = Config(feature_type='i3d', video_paths=[vid], cpu=True,
                 stack_size=10, step_size=10, flow_type='pwc')
    ex = ExtractI3D(cfg, external_call=True)
    out = ex(torch.arange(1))[0]
    assert out['rgb'].shape == (2, 1024)
    assert out['flow'].shape == (2, 1024)
    assert np.isfinite(out['rgb']).all() and np.isfinite(out['flow']).all()


def test_i3d_short_video_resampled(tmp_path):
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.i3d import ExtractI3D
    from video_features_amd.io.y4m import write_y4m
    frames = synthetic_frames(t=6, h=64, w=64)   # shorter than stack+1
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, frames, fps=25.0)
    cfg = Config(feature_type='i3d', video_paths=[vid], cpu=True,
                 stack_size=10, step_size=10, streams=['rgb'])
    ex = ExtractI3D(cfg, external_call=True)
    out = ex(torch.arange(1))[0]
    assert out['rgb'].shape == (1, 1024)


def test_mel_frontend_shapes():
    from video_features_amd.models.vggish import waveform_to_examples
    torch.manual_seed(0)
    wav = torch.randn(16000 * 2)   # 2 s
    ex = waveform_to_examples(wav)
    assert ex.shape == (2, 96, 64)
    assert torch.isfinite(ex).all()


def test_mel_frontend_tone_peak():
    # a 1 kHz tone must put its energy in the right mel band
    import numpy as np
    from video_features_amd.models.vggish import (MEL_MAX_HZ, MEL_MIN_HZ,
                                                  waveform_to_examples)
    t = np.arange(16000) / 16000
    wav = torch.from_numpy(np.sin(2 * np.pi * 1000 * t).astype(np.float32))
    ex = waveform_to_examples(wav)[0]     # (96, 64)
    band = ex.mean(0).argmax().item()
    # HTK mel position of 1 kHz within [125, 7500]
    def mel(f):
        return 1127 * np.log(1 + f / 700)
    frac = (mel(1000) - mel(MEL_MIN_HZ)) / (mel(MEL_MAX_HZ) - mel(MEL_MIN_HZ))
    assert abs(band - frac * 64) < 4, (band, frac * 64)


def test_vggish_net_shapes():
    from video_features_amd.models.vggish import VGGish
    torch.manual_seed(0)
    m = VGGish().eval()
    n = sum(p.numel() for p in m.parameters())
    assert 60e6 < n < 80e6, n   # published VGGish ≈ 72M params
    with torch.no_grad():
        out = m(torch.randn(3, 96, 64))
    assert out.shape == (3, 128)


def test_vggish_extractor_wav_input(tmp_path):
    from video_features_amd.extractors.vggish import ExtractVGGish
    from video_features_amd.io.audio import write_wav
    t = np.arange(16000 * 2) / 16000
    sig = (0.3 * np.sin(2 * np.pi * 440 * t)).astype(np.float32)
    wav = str(tmp_path / 'a.wav')
    write_wav(wav, sig, 16000)
    cfg = Config(feature_type='vggish', video_paths=[wav], cpu=True)
    ex = ExtractVGGish(cfg, external_call=True)
    out = ex(torch.arange(1))[0]
    assert out['vggish'].shape == (2, 128)
    assert np.isfinite(out['vggish']).all()


def test_vggish_sidecar_wav(tmp_path, frames16):
    # a video with a sidecar .wav uses it without ffmpeg
    from video_features_amd.extractors.vggish import ExtractVGGish
    from video_features_amd.io.audio import write_wav
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, frames16, fps=25.0)
    sig = np.random.default_rng(0).standard_normal(16000).astype(np.float32) * 0.1
    write_wav(str(tmp_path / 'v.wav'), sig, 16000)
    cfg = Config(feature_type='vggish_torch', video_paths=[vid], cpu=True)
    ex = ExtractVGGish(cfg, external_call=True)
    out = ex(torch.arange(1))[0]
    assert out['vggish_torch'].shape == (1, 128)
