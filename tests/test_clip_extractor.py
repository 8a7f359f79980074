import os

import numpy as np
import pytest
import torch

from video_features_amd.config import Config
from video_features_amd.models.clip_vit import ViTConfig, VisionTransformer


def tiny_vit():
    return VisionTransformer(ViTConfig(input_resolution=64, patch_size=16,
                                       width=64, layers=2, heads=4,
                                       output_dim=32))


def test_vit_shapes():
    torch.manual_seed(0)
    m = tiny_vit().eval()
    x = torch.randn(3, 3, 64, 64)
    with torch.no_grad():
        out = m.encode_image(x)
    assert out.shape == (3, 32)
    assert torch.isfinite(out).all()


def test_vit_attention_matches_sdpa():
    # our ops.attention torch path vs torch's scaled_dot_product_attention
    from video_features_amd import ops
    torch.manual_seed(1)
    q, k, v = [torch.randn(2, 4, 10, 16) for _ in range(3)]
    ours = ops.attention(q, k, v)
    ref = torch.nn.functional.scaled_dot_product_attention(q, k, v)
    assert torch.allclose(ours, ref, atol=1e-5)


def test_clip_extractor_end_to_end(tmp_path, y4m_video):
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[y4m_video],
                 extract_method='uni_4', cpu=True, on_extraction='save_numpy',
                 output_path=str(tmp_path / 'out'), tmp_path=str(tmp_path / 'tmp'))
    from video_features_amd.extractors.clip import ExtractCLIP
    ex = ExtractCLIP(cfg)
    ex(torch.arange(1))
    out_dir = os.path.join(str(tmp_path / 'out'), 'CLIP-ViT-B/32')
    files = os.listdir(out_dir)
    assert 'vid_CLIP-ViT-B_32.npy' in files
    feats = np.load(os.path.join(out_dir, 'vid_CLIP-ViT-B_32.npy'))
    assert feats.shape == (4, 512)
    assert np.isfinite(feats).all()


def test_clip_external_call(y4m_video):
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[y4m_video],
                 extract_method='uni_3', cpu=True)
    from video_features_amd.extractors.clip import ExtractCLIP
    ex = ExtractCLIP(cfg, external_call=True)
    feats_list = ex(torch.arange(1))
    assert len(feats_list) == 1
    fd = feats_list[0]
    assert fd['CLIP-ViT-B/32'].shape == (3, 512)
    assert 'fps' in fd and 'timestamps_ms' in fd
    assert len(fd['timestamps_ms']) == 3


def test_error_isolation(tmp_path, y4m_video, capsys):
    # a corrupt video must not kill the shard (reference extract_clip.py:70-84)
    bad = tmp_path / 'bad.y4m'
    bad.write_bytes(b'garbage not a y4m file at all')
    cfg = Config(feature_type='CLIP-ViT-B/32',
                 video_paths=[str(bad), y4m_video],
                 extract_method='uni_2', cpu=True)
    from video_features_amd.extractors.clip import ExtractCLIP
    ex = ExtractCLIP(cfg, external_call=True)
    feats_list = ex(torch.arange(2))
    assert len(feats_list) == 1      # good video still extracted
    assert 'Extraction failed' in capsys.readouterr().out


def test_clip_golden_regression(tmp_path):
    """Golden-feature regression: fixed seed-0 weights + fixed
    synthetic video must reproduce the recorded features.  Guards silent
    numerics drift in the preprocess/sampler/model chain."""
    import json
    import os
    from tests.conftest import synthetic_frames
    from video_features_amd.io.y4m import write_y4m
    from video_features_amd.extractors.clip import ExtractCLIP
    gold = json.load(open(os.path.join(os.path.dirname(__file__),
                                       'golden_clip.json')))
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, synthetic_frames(t=24, h=64, w=96, seed=3), fps=25.0)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[vid], cpu=True,
                 extract_method='uni_6', seed=0)
    out = ExtractCLIP(cfg, external_call=True)(torch.arange(1))[0]
    f = out['CLIP-ViT-B/32']
    assert list(f.shape) == gold['shape']
    np.testing.assert_allclose(f[0, :8], gold['first_row_8'],
                               atol=2e-3, rtol=1e-3)
    assert abs(float(f.mean()) - gold['mean']) < 1e-3
    assert abs(float(f.std()) - gold['std']) < 1e-3
    np.testing.assert_allclose(out['timestamps_ms'], gold['timestamps_ms'])


def test_decode_pipeline_matches_serial(tmp_path):
    """The one-video-ahead decode thread must not change results (multi-
    video run vs per-video serial runs)."""
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    paths = []
    for i in range(3):
        p = str(tmp_path / f'v{i}.y4m')
        write_y4m(p, synthetic_frames(t=10 + i, h=64, w=64, seed=i), fps=25.0)
        paths.append(p)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=paths, cpu=True,
                 extract_method='uni_3', seed=0)
    ex = ExtractCLIP(cfg, external_call=True)
    piped = ex(torch.arange(3))
    for i, p in enumerate(paths):
        solo = ExtractCLIP(cfg.replace(video_paths=[p]),
                           external_call=True)(torch.arange(1))[0]
        np.testing.assert_allclose(piped[i]['CLIP-ViT-B/32'],
                                   solo['CLIP-ViT-B/32'], rtol=1e-5,
                                   atol=1e-6)


def test_decode_pipeline_error_isolation(tmp_path):
    """A video whose prefetched decode fails must be skipped without
    poisoning the next video's result."""
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    good1 = str(tmp_path / 'a.y4m')
    bad = str(tmp_path / 'b.y4m')
    good2 = str(tmp_path / 'c.y4m')
    write_y4m(good1, synthetic_frames(t=8, h=64, w=64), fps=25.0)
    open(bad, 'wb').write(b'NOT A VIDEO AT ALL' * 10)
    write_y4m(good2, synthetic_frames(t=8, h=64, w=64, seed=2), fps=25.0)
    cfg = Config(feature_type='CLIP-ViT-B/32',
                 video_paths=[good1, bad, good2], cpu=True,
                 extract_method='uni_2')
    out = ExtractCLIP(cfg, external_call=True)(torch.arange(3))
    assert len(out) == 2   # bad one skipped, both good ones extracted
    for fd in out:
        assert fd['CLIP-ViT-B/32'].shape == (2, 512)


def test_clip_rn50_extractor(tmp_path):
    """CLIP-RN50: the ModifiedResNet tower the reference codes but never
    exposes — here a first-class feature type, (N, 1024) features."""
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, synthetic_frames(t=10, h=64, w=64), fps=25.0)
    cfg = Config(feature_type='CLIP-RN50', video_paths=[vid], cpu=True,
                 extract_method='uni_2')
    out = ExtractCLIP(cfg, external_call=True)(torch.arange(1))[0]
    assert out['CLIP-RN50'].shape == (2, 1024)
    assert np.isfinite(out['CLIP-RN50']).all()


def test_clip_rn50_openai_scheme_roundtrip():
    """OpenAI RN50 visual-tower key scheme loads via the converter."""
    from video_features_amd.models.clip_resnet import build_clip_resnet
    from video_features_amd.utils.convert_checkpoints import convert_auto
    torch.manual_seed(0)
    m = build_clip_resnet('CLIP-RN50')
    sd = m.state_dict()
    legacy = {'visual.' + k: v.clone() for k, v in sd.items()}
    legacy['logit_scale'] = torch.zeros(())
    m2 = build_clip_resnet('CLIP-RN50')
    m2.load_state_dict(convert_auto(legacy))
    for k, v in m2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_resume_skip_does_not_mix_prefetched_decodes(tmp_path):
    """Regression: with --resume skipping a middle video, the prefetched
    decode of the skipped video must NOT be consumed by the next one."""
    import os
    paths = []
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    for i in range(3):
        p = str(tmp_path / f'v{i}.y4m')
        write_y4m(p, synthetic_frames(t=10, h=64, w=64, seed=10 + i),
                  fps=25.0)
        paths.append(p)
    out_dir = tmp_path / 'out'
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=paths, cpu=True,
                 extract_method='uni_2', seed=0, resume=True,
                 on_extraction='save_numpy', output_path=str(out_dir),
                 tmp_path=str(tmp_path / 'tmp'))
    # pre-extract ONLY the middle video so resume skips it
    ExtractCLIP(cfg.replace(video_paths=[paths[1]]))(torch.arange(1))
    ExtractCLIP(cfg)(torch.arange(3))
    # ground truth per video, solo runs
    for i, p in enumerate(paths):
        solo = ExtractCLIP(cfg.replace(video_paths=[p],
                                       on_extraction='print'),
                           external_call=True)(torch.arange(1))[0]
        feat_dir = out_dir / 'CLIP-ViT-B/32'
        got = np.load(feat_dir / f'v{i}_CLIP-ViT-B_32.npy')
        np.testing.assert_allclose(got, solo['CLIP-ViT-B/32'],
                                   rtol=1e-5, atol=1e-6, err_msg=p)


def test_clip4clip_auto_checkpoint(tmp_path, monkeypatch, y4m_video):
    """CLIP4CLIP auto-loads its conventional checkpoint when --weights_path
    is absent (reference extract_clip.py:58-63 behavior)."""
    import os
    import torch
    from video_features_amd.config import Config
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.models.clip_vit import VisionTransformer
    torch.manual_seed(7)
    m = VisionTransformer()
    cache = tmp_path / 'wcache'
    cache.mkdir()
    torch.save(m.state_dict(), str(cache / 'CLIP4CLIP-ViT-B-32.pth'))
    monkeypatch.setenv('VFA_WEIGHTS_CACHE', str(cache))
    cfg = Config(feature_type='CLIP4CLIP-ViT-B-32', video_paths=[y4m_video],
                 cpu=True, tmp_path=str(tmp_path / 'tmp'))
    ex = ExtractCLIP(cfg, external_call=True)
    built = ex.models_for(torch.device('cpu'))
    for k, v in built.state_dict().items():
        assert torch.equal(v, m.state_dict()[k]), k
