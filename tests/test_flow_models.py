import numpy as np
import pytest
import torch

from video_features_amd.config import Config


def test_pwc_correlation_torch_matches_naive():
    from video_features_amd.ops import _pwc_correlation_torch
    torch.manual_seed(0)
    f1 = torch.randn(2, 8, 6, 7)
    f2 = torch.randn(2, 8, 6, 7)
    out = _pwc_correlation_torch(f1, f2, max_disp=2)
    assert out.shape == (2, 25, 6, 7)
    # naive check at one pixel / displacement: d=(dy=+1, dx=-1) → index
    dy, dx = 1, -1
    ch = (dy + 2) * 5 + (dx + 2)
    y, x = 3, 4
    expected = (f1[0, :, y, x] * f2[0, :, y + dy, x + dx]).mean()
    assert torch.allclose(out[0, ch, y, x], expected, atol=1e-5)
    # out-of-bounds displacements read zero padding
    ch0 = 0  # dy=-2, dx=-2 at pixel (0,0) → fully out of bounds
    assert out[0, ch0, 0, 0] == 0


def test_bilinear_warp_identity_and_shift():
    from video_features_amd.ops import bilinear_warp
    x = torch.arange(24.0).reshape(1, 1, 4, 6)
    zero = torch.zeros(1, 2, 4, 6)
    out = bilinear_warp(x, zero)
    assert torch.allclose(out, x)
    # shift right by 1: output[y, x] = input[y, x+1]
    flow = torch.zeros(1, 2, 4, 6)
    flow[:, 0] = 1.0
    out = bilinear_warp(x, flow)
    assert torch.allclose(out[0, 0, :, :-1], x[0, 0, :, 1:])


def test_grid_sample_bilinear_pixel_coords():
    from video_features_amd.ops import grid_sample_bilinear
    x = torch.arange(12.0).reshape(1, 1, 3, 4)
    coords = torch.tensor([[[[1.0, 1.0], [2.5, 0.0]]]])   # (B, 1, 2, 2[xy])
    out = grid_sample_bilinear(x, coords)
    assert out.shape == (1, 1, 1, 2)
    assert torch.allclose(out[0, 0, 0, 0], torch.tensor(5.0))   # (y=1,x=1)
    assert torch.allclose(out[0, 0, 0, 1], torch.tensor(2.5))   # interp x=2.5


def test_raft_forward_shapes():
    from video_features_amd.models.raft import RAFT
    torch.manual_seed(0)
    m = RAFT(iters=2).eval()
    with torch.no_grad():
        flow = m(torch.rand(1, 3, 64, 96) * 255, torch.rand(1, 3, 64, 96) * 255)
    assert flow.shape == (1, 2, 64, 96)
    assert torch.isfinite(flow).all()


def test_raft_padder():
    from video_features_amd.models.raft import InputPadder
    x = torch.zeros(1, 3, 60, 90)
    p = InputPadder(x.shape)
    (xp,) = p.pad(x)
    assert xp.shape[-2] % 8 == 0 and xp.shape[-1] % 8 == 0
    assert p.unpad(xp).shape == x.shape


def test_pwc_forward_shapes():
    from video_features_amd.models.pwc import PWCNet
    torch.manual_seed(0)
    m = PWCNet().eval()
    with torch.no_grad():
        flow = m(torch.rand(1, 3, 64, 96) * 255, torch.rand(1, 3, 64, 96) * 255)
    assert flow.shape == (1, 2, 64, 96)
    assert torch.isfinite(flow).all()


def test_flow_viz():
    from video_features_amd.utils.flow_viz import flow_to_image, make_colorwheel
    assert make_colorwheel().shape == (55, 3)
    img = flow_to_image(np.random.randn(8, 8, 2).astype(np.float32))
    assert img.shape == (8, 8, 3) and img.dtype == np.uint8


@pytest.mark.parametrize('feature_type', ['raft', 'pwc'])
def test_flow_extractor_end_to_end(y4m_video, feature_type):
    from video_features_amd.models import raft as raft_mod
    cfg = Config(feature_type=feature_type, video_paths=[y4m_video],
                 batch_size=4, cpu=True)
    from video_features_amd.models.registry import get_extractor_class
    cls = get_extractor_class(feature_type)
    ex = cls(cfg, external_call=True)
    # shrink RAFT iterations for CPU test speed
    if feature_type == 'raft':
        models = ex.models_for(torch.device('cpu'))
        models.iters = 2
    out = ex(torch.arange(1))[0]
    flow = out[feature_type]
    assert flow.shape == (15, 2, 64, 96)   # T-1 flow frames
    assert np.isfinite(flow).all()


def test_raft_checkpoint_compat_separate_zr():
    """Reference-style checkpoints with separate convz/convr keys load into
    the merged convzr layout."""
    from video_features_amd.models.raft import SepConvGRU
    torch.manual_seed(0)
    gru = SepConvGRU(8, 16)
    sd = gru.state_dict()
    legacy = {}
    for i in ('1', '2'):
        w = sd[f'convzr{i}.weight']
        bi = sd[f'convzr{i}.bias']
        legacy[f'convz{i}.weight'], legacy[f'convr{i}.weight'] = \
            w[:8].clone(), w[8:].clone()
        legacy[f'convz{i}.bias'], legacy[f'convr{i}.bias'] = \
            bi[:8].clone(), bi[8:].clone()
        legacy[f'convq{i}.weight'] = sd[f'convq{i}.weight'].clone()
        legacy[f'convq{i}.bias'] = sd[f'convq{i}.bias'].clone()
    gru2 = SepConvGRU(8, 16)
    gru2.load_state_dict(legacy)
    for k, v in gru2.state_dict().items():
        assert torch.equal(v, sd[k]), k


def test_show_pred_saves_flow_visualization(tmp_path):
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.pwc import ExtractPWC
    from video_features_amd.io.y4m import write_y4m
    import os
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, synthetic_frames(t=5, h=64, w=64), fps=25.0)
    cfg = Config(feature_type='pwc', video_paths=[vid], cpu=True,
                 show_pred=True, tmp_path=str(tmp_path / 'tmp'))
    out = ExtractPWC(cfg, external_call=True)(torch.arange(1))[0]
    assert out['pwc'].shape[0] == 4
    ppms = [f for f in os.listdir(tmp_path / 'tmp') if f.endswith('.ppm')]
    assert ppms, 'flow visualization not saved'
