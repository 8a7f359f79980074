"""GPU end-to-end extractor runs: decode → sample → preprocess → native
GPU model → features, for every family, on real (synthetic) video files.
This is the actual user path (`indices.device = cuda:0`), so it also
verifies the HIP extension is the code that runs on a GPU box."""
import numpy as np
import pytest
import torch

from tests.conftest import synthetic_frames
from video_features_amd.config import Config
from video_features_amd.io.y4m import write_y4m

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def _vid(tmp_path, t=40, h=96, w=128):
    p = str(tmp_path / 'v.y4m')
    write_y4m(p, synthetic_frames(t=t, h=h, w=w), fps=25.0)
    return p


def _run(cfg, dev):
    from video_features_amd.models.registry import get_extractor_class
    ex = get_extractor_class(cfg.feature_type)(cfg, external_call=True)
    out = ex(torch.arange(1, device=dev))
    assert len(out) == 1, 'extraction failed (error swallowed per-video)'
    return out[0]


def test_clip_e2e(dev, tmp_path):
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[_vid(tmp_path)],
                 extract_method='uni_8', tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    f = out['CLIP-ViT-B/32']
    assert f.shape == (8, 512) and np.isfinite(f).all()
    assert out['timestamps_ms'].shape == (8,)


def test_i3d_raft_e2e(dev, tmp_path):
    cfg = Config(feature_type='i3d', video_paths=[_vid(tmp_path, t=70)],
                 flow_type='raft', batch_size=2,
                 tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['rgb'].shape == (1, 1024) and out['flow'].shape == (1, 1024)
    assert np.isfinite(out['rgb']).all() and np.isfinite(out['flow']).all()


def test_i3d_pwc_e2e(dev, tmp_path):
    cfg = Config(feature_type='i3d', video_paths=[_vid(tmp_path, t=70)],
                 flow_type='pwc', tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['rgb'].shape == (1, 1024) and out['flow'].shape == (1, 1024)


def test_resnet50_e2e(dev, tmp_path):
    cfg = Config(feature_type='resnet50', video_paths=[_vid(tmp_path)],
                 batch_size=16, tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['resnet50'].shape == (40, 2048)
    assert np.isfinite(out['resnet50']).all()


def test_r21d_e2e(dev, tmp_path):
    cfg = Config(feature_type='r21d_rgb', video_paths=[_vid(tmp_path)],
                 tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['r21d_rgb'].shape == (2, 512)   # 40 frames / 16-stack


def test_raft_flow_e2e(dev, tmp_path):
    cfg = Config(feature_type='raft', video_paths=[_vid(tmp_path, t=9)],
                 batch_size=4, tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['raft'].shape == (8, 2, 96, 128)
    assert np.isfinite(out['raft']).all()


def test_pwc_flow_e2e(dev, tmp_path):
    cfg = Config(feature_type='pwc', video_paths=[_vid(tmp_path, t=9)],
                 batch_size=4, tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['pwc'].shape == (8, 2, 96, 128)


def test_vggish_e2e(dev, tmp_path):
    from video_features_amd.io.audio import write_wav
    vid = _vid(tmp_path, t=50)
    sig = np.random.default_rng(0).standard_normal(16000 * 2) \
        .astype(np.float32) * 0.1
    write_wav(str(tmp_path / 'v.wav'), sig, 16000)
    cfg = Config(feature_type='vggish_torch', video_paths=[vid],
                 tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    assert out['vggish_torch'].shape == (2, 128)


def test_clip_rn50_e2e(dev, tmp_path):
    cfg = Config(feature_type='CLIP-RN50', video_paths=[_vid(tmp_path)],
                 extract_method='uni_4', tmp_path=str(tmp_path / 'tmp'))
    out = _run(cfg, dev)
    f = out['CLIP-RN50']
    assert f.shape == (4, 1024) and np.isfinite(f).all()
