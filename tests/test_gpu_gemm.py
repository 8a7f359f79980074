"""GPU numerics for the fused MFMA linear kernel (ops.linear_act) vs the
plain torch fp32 reference."""
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


def _ref(x, w, b, act):
    y = torch.nn.functional.linear(x.float(), w.float(),
                                   b.float() if b is not None else None)
    if act == 'relu':
        return y.relu()
    if act == 'quick_gelu':
        return y * torch.sigmoid(1.702 * y)
    if act == 'gelu':
        return torch.nn.functional.gelu(y, approximate='tanh')
    return y


@pytest.mark.parametrize('m,n,k', [(256, 128, 64), (512, 384, 192),
                                   (1000, 768, 3072), (9600, 3072, 768),
                                   (129, 256, 128), (333, 100, 72),
                                   (2048, 2048, 520), (770, 530, 336)])
@pytest.mark.parametrize('act', ['none', 'relu', 'quick_gelu', 'gelu'])
def test_linear_act(dev, m, n, k, act):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = (torch.randn(m, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    w = (torch.randn(n, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, b, act)
    assert out.shape == (m, n) and out.dtype == torch.bfloat16
    ref = _ref(x, w, b, act)
    err = (out.float() - ref).abs()
    rel = err.max().item() / max(ref.abs().max().item(), 1e-6)
    assert rel < 3e-2, rel
    # tight mean check: accumulation is fp32, only I/O is bf16
    assert (err.mean() / ref.abs().mean().clamp_min(1e-6)).item() < 5e-3


@pytest.mark.parametrize('m,n,k', [
    (131072, 64, 64),     # resnet l1 conv1 regime (thin-K streaming path)
    (131072, 256, 64),    # l1 conv3
    (131072, 128, 256),   # l2 conv1
    (70000, 512, 128),    # l2 conv3 (ragged M)
    (65552, 48, 96),      # ragged N + M just over the threshold
    (131072, 192, 144),   # r21d temporal conv regime: K%32 != 0 (masked
                          # tail fragment + zero-filled W rows)
    (131072, 256, 232),   # K%32 = 8
])
@pytest.mark.parametrize('act', ['none', 'relu'])
def test_linear_thin_streaming(dev, m, n, k, act):
    """The M-huge / K-shallow streaming kernel (1x1-conv-as-GEMM shapes)."""
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = (torch.randn(m, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    w = (torch.randn(n, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, b, act)
    ref = _ref(x, w, b, act)
    err = (out.float() - ref).abs()
    rel = err.max().item() / max(ref.abs().max().item(), 1e-6)
    assert rel < 3e-2, rel


@pytest.mark.parametrize('m,n,k', [
    (19200, 2304, 768),   # CLIP qkv (fb384) — the 8-phase pipelined path
    (19200, 768, 3072),   # CLIP fc2
    (4096, 2048, 768),
    (4096, 256, 192),     # minimum K depth (3 tiles)
])
@pytest.mark.parametrize('act', ['none', 'quick_gelu'])
def test_linear_8phase(dev, m, n, k, act):
    """The deep-pipelined 256^2 kernel (full tiles, counted-vmcnt raw
    barriers); numerics vs fp32 torch."""
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = (torch.randn(m, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    w = (torch.randn(n, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, b, act)
    ref = _ref(x, w, b, act)
    err = (out.float() - ref).abs()
    rel = err.max().item() / max(ref.abs().max().item(), 1e-6)
    assert rel < 3e-2, rel
    assert (err.mean() / ref.abs().mean().clamp_min(1e-6)).item() < 5e-3


def test_linear_8phase_residual(dev):
    """fc2 + residual add fused in the 8-phase epilogue."""
    ops = _hip_loaded()
    torch.manual_seed(4)
    m, n, k = 19200, 768, 3072
    x = (torch.randn(m, k, device=dev) / 8).to(torch.bfloat16)
    w = (torch.randn(n, k, device=dev) / 8).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    r = torch.randn(m, n, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, b, 'none', r)
    ref = _ref(x, w, b, 'none') + r.float()
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert rel < 3e-2, rel


def test_linear_thin_residual(dev):
    """bottleneck conv3 epilogue: bias + residual + relu on the thin path."""
    ops = _hip_loaded()
    torch.manual_seed(3)
    m, n, k = 131072, 256, 64
    x = (torch.randn(m, k, device=dev) / 2).to(torch.bfloat16)
    w = (torch.randn(n, k, device=dev) / 2).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    r = torch.randn(m, n, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, b, 'relu', r)
    ref = (_ref(x, w, b, 'none') + r.float()).relu()
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert rel < 3e-2, rel


def test_linear_act_no_bias(dev):
    ops = _hip_loaded()
    torch.manual_seed(1)
    x = torch.randn(384, 256, device=dev).to(torch.bfloat16)
    w = torch.randn(128, 256, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, None, 'none')
    ref = _ref(x, w, None, 'none')
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert rel < 3e-2, rel


def test_linear_act_3d_input(dev):
    ops = _hip_loaded()
    torch.manual_seed(2)
    x = torch.randn(4, 50, 768, device=dev).to(torch.bfloat16)
    w = torch.randn(3072, 768, device=dev).to(torch.bfloat16)
    b = torch.zeros(3072, device=dev).to(torch.bfloat16)
    out = ops.linear_act(x, w, b, 'quick_gelu')
    assert out.shape == (4, 50, 3072)
    ref = _ref(x.reshape(-1, 768), w, b, 'quick_gelu').reshape(4, 50, 3072)
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert rel < 3e-2, rel
