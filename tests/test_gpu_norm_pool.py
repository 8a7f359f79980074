"""GPU numerics: fused InstanceNorm2d(+ReLU) and TF-SAME max_pool3d vs
plain PyTorch fp32 references."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def _hip_loaded():
    from video_features_amd import ops
    assert ops.hip_available()
    return ops


@pytest.mark.parametrize('nhwc', [False, True])
@pytest.mark.parametrize('relu', [False, True])
@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-4),
                                       (torch.bfloat16, 5e-2)])
def test_instance_norm(dev, nhwc, relu, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    b, c, h, w = 5, 96, 37, 53
    mf = torch.channels_last if nhwc else torch.contiguous_format
    x = (torch.randn(b, c, h, w, device=dev) * 3 + 1).to(dtype) \
        .contiguous(memory_format=mf)
    out = ops.instance_norm(x, relu=relu, nhwc=nhwc)
    ref = torch.nn.functional.instance_norm(x.float())
    if relu:
        ref = ref.relu()
    assert (out.float() - ref).abs().max().item() < tol
    # layout preserved
    assert out.is_contiguous(memory_format=mf)


@pytest.mark.parametrize('kernel,stride', [((1, 3, 3), (1, 2, 2)),
                                           ((3, 3, 3), (2, 2, 2)),
                                           ((2, 2, 2), (2, 2, 2)),
                                           ((3, 3, 3), (1, 1, 1))])
@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_maxpool3d_same(dev, kernel, stride, dtype):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = torch.randn(2, 7, 9, 29, 31, device=dev).to(dtype)
    out = ops.maxpool3d_same(x, kernel, stride)
    # reference: explicit TF-SAME pad (zeros) + torch max_pool3d
    import os
    os.environ['VFA_FORCE_TORCH_OPS'] = '1'
    try:
        ref = ops.maxpool3d_same(x, kernel, stride)
    finally:
        del os.environ['VFA_FORCE_TORCH_OPS']
    assert out.shape == ref.shape
    assert torch.equal(out.float(), ref.float())


def test_maxpool3d_same_matches_i3d_geometry(dev):
    """I3D pool geometry: out dims = ceil(in/stride) per TF-SAME."""
    ops = _hip_loaded()
    x = torch.randn(1, 4, 64, 224, 224, device=dev, dtype=torch.bfloat16)
    out = ops.maxpool3d_same(x, (1, 3, 3), (1, 2, 2))
    assert out.shape == (1, 4, 64, 112, 112)
    out = ops.maxpool3d_same(x, (3, 3, 3), (2, 2, 2))
    assert out.shape == (1, 4, 32, 112, 112)


@pytest.mark.parametrize('kt,st,p0,t', [(3, 1, 1, 8), (3, 2, 1, 16),
                                        (3, 2, 1, 15)])
def test_temporal_merge_fused(dev, kt, st, p0, t):
    """Fused temporal tap merge vs the strided-add torch composition."""
    from video_features_amd.models import _flat3d
    torch.manual_seed(0)
    b, o, h, w = 2, 24, 5, 7
    y = torch.randn(b * t, kt * o, h, w, device=dev) \
        .contiguous(memory_format=torch.channels_last)
    out = _flat3d.temporal_merge(y, b, kt, st, p0)      # fused on GPU
    ref = _flat3d.temporal_merge(y.cpu().float(), b, kt, st, p0)
    assert out.shape == ref.shape
    assert torch.allclose(out.cpu().float(), ref, atol=1e-5), \
        (out.cpu().float() - ref).abs().max().item()
