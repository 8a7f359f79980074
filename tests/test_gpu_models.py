"""GPU forwards of the non-CLIP model families vs their CPU fp32
references (same weights, same inputs)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def _cos(a, b):
    return torch.nn.functional.cosine_similarity(
        a.flatten().float(), b.flatten().float(), dim=0).item()


def test_r21d_gpu_vs_cpu(dev):
    from video_features_amd.models.r21d import R2Plus1D18
    torch.manual_seed(0)
    m = R2Plus1D18().eval()
    x = torch.randn(2, 3, 16, 112, 112)
    with torch.no_grad():
        ref = m.forward_features(x)
        out = m.to(dev).forward_features(x.to(dev)).cpu()
    assert out.shape == (2, 512)
    assert _cos(out, ref) > 0.999


def test_r21d_gpu_bf16(dev):
    from video_features_amd.models.r21d import R2Plus1D18
    torch.manual_seed(0)
    m = R2Plus1D18().eval()
    x = torch.randn(2, 3, 16, 112, 112)
    with torch.no_grad():
        ref = m.forward_features(x)
        out = m.to(dev, torch.bfloat16) \
            .forward_features(x.to(dev, torch.bfloat16)).float().cpu()
    assert _cos(out, ref) > 0.98


def test_i3d_gpu_vs_cpu(dev):
    from video_features_amd.models.i3d import I3D
    torch.manual_seed(0)
    m = I3D(modality='rgb').eval()
    x = torch.randn(1, 3, 16, 128, 128)
    with torch.no_grad():
        ref = m.forward_features(x)
        out = m.to(dev).forward_features(x.to(dev)).cpu()
    assert out.shape == (1, 1024)
    assert _cos(out, ref) > 0.999


@pytest.mark.parametrize('modality,t', [('rgb', 16), ('flow', 15)])
def test_i3d_gpu_bf16_flat_stem(dev, modality, t):
    """bf16 GPU path: the 7x7x7 stem runs as the channel-padded 7-tap
    merged conv2d + strided temporal merge (no conv3d at all); odd T
    exercises the asymmetric TF-SAME temporal pad."""
    from video_features_amd.models.i3d import I3D
    torch.manual_seed(0)
    c = 3 if modality == 'rgb' else 2
    m = I3D(modality=modality).eval()
    x = torch.randn(1, c, t, 128, 128)
    with torch.no_grad():
        ref = m.forward_features(x)
        out = m.to(dev).to(torch.bfloat16).forward_features(
            x.to(dev).to(torch.bfloat16)).float().cpu()
    assert out.shape == (1, 1024)
    assert _cos(out, ref) > 0.99


def test_resnet50_gpu_vs_cpu(dev):
    from video_features_amd.models.resnet import build_resnet
    torch.manual_seed(0)
    m = build_resnet('resnet50').eval()
    x = torch.randn(4, 3, 224, 224)
    with torch.no_grad():
        ref = m.forward_features(x)
        mm = m.to(dev).to(memory_format=torch.channels_last)
        out = mm.forward_features(
            x.to(dev).contiguous(memory_format=torch.channels_last)).cpu()
    assert out.shape == (4, 2048)
    assert _cos(out, ref) > 0.999


def test_vggish_gpu_vs_cpu(dev):
    from video_features_amd.models.vggish import VGGish, waveform_to_examples
    torch.manual_seed(0)
    m = VGGish().eval()
    wav = torch.sin(torch.arange(16000 * 2) / 16000 * 2 * 3.14159 * 440)
    ex = waveform_to_examples(wav)
    with torch.no_grad():
        ref = m(ex)
        out = m.to(dev)(ex.to(dev)).cpu()
    assert out.shape[1] == 128
    assert _cos(out, ref) > 0.999


def test_pwc_gpu_vs_cpu(dev):
    from video_features_amd.models.pwc import PWCNet
    torch.manual_seed(0)
    m = PWCNet().eval()
    x1 = torch.rand(2, 3, 128, 192) * 255
    x2 = torch.rand(2, 3, 128, 192) * 255
    with torch.no_grad():
        ref = m(x1, x2)
        out = m.to(dev)(x1.to(dev), x2.to(dev)).cpu()
    assert out.shape == ref.shape == (2, 2, 128, 192)
    assert _cos(out, ref) > 0.99


def test_resnet50_fused_bottleneck_path(dev):
    """BN-folded bf16 CL path (1x1 convs as fused MFMA GEMMs with
    residual+ReLU epilogue) vs the folded CPU fp32 reference."""
    from video_features_amd.models.resnet import build_resnet
    from video_features_amd.utils.fold_bn import fold_batchnorms
    torch.manual_seed(0)
    m = build_resnet('resnet50').eval()
    for mod in m.modules():
        if isinstance(mod, torch.nn.BatchNorm2d):
            mod.running_mean.normal_(0, 0.2)
            mod.running_var.uniform_(0.5, 2.0)
    fold_batchnorms(m)
    x = torch.randn(4, 3, 128, 128)
    with torch.no_grad():
        ref = m.forward_features(x)
        mm = m.to(dev, torch.bfloat16).to(memory_format=torch.channels_last)
        out = mm.forward_features(
            x.to(dev, torch.bfloat16)
            .contiguous(memory_format=torch.channels_last)).float().cpu()
    assert _cos(out, ref) > 0.99, _cos(out, ref)
