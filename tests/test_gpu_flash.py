"""GPU tests for the MFMA flash attention path (head_dim 64)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def test_mfma_fragment_layout(dev):
    """Verify the assumed v_mfma_f32_16x16x32_bf16 lane layouts: D = A @ B
    with ASYMMETRIC operands (catches transposes — guide G9)."""
    from video_features_amd.ops import _vfa_hip
    torch.manual_seed(0)
    a = torch.randn(16, 32, device=dev)
    b = torch.randn(32, 16, device=dev)
    # quantize to bf16 so the reference matches the MFMA inputs exactly
    a = a.to(torch.bfloat16).float()
    b = b.to(torch.bfloat16).float()
    d = _vfa_hip.mfma_gemm16(a, b)
    ref = a @ b
    err = (d - ref).abs().max().item()
    assert err < 1e-2, f'MFMA layout mismatch, max err {err}'


@pytest.mark.parametrize('n,h', [(50, 12), (64, 12), (197, 12), (49, 4),
                                 (130, 8)])
def test_flash_qkv_vs_fp32(dev, n, h):
    from video_features_amd import ops
    torch.manual_seed(0)
    b, d = 3, 64
    e = h * d
    qkv = torch.randn(b, n, 3 * e, device=dev, dtype=torch.bfloat16)
    out = ops.mhsa_fused(qkv, h).float()
    # fp32 reference from the same qkv values
    q, k, v = qkv.float().view(b, n, 3, h, d).permute(2, 0, 3, 1, 4).unbind(0)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q, k, v, scale=1.0 / d ** 0.5)
    ref = ref.permute(0, 2, 1, 3).reshape(b, n, e)
    err = (out - ref).abs().max().item()
    assert err < 3e-2, (n, h, err)


def test_flash_qkv_softmax_stability(dev):
    # large-magnitude scores exercise the online-softmax rescale path
    from video_features_amd import ops
    torch.manual_seed(1)
    b, n, h, d = 2, 197, 4, 64
    qkv = (torch.randn(b, n, 3 * h * d, device=dev, dtype=torch.bfloat16) * 8)
    out = ops.mhsa_fused(qkv, h).float()
    assert torch.isfinite(out).all()
    q, k, v = qkv.float().view(b, n, 3, h, d).permute(2, 0, 3, 1, 4).unbind(0)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q, k, v, scale=1.0 / d ** 0.5).permute(0, 2, 1, 3).reshape(b, n, h * d)
    # inputs are x8 scaled → compare relative to output magnitude (bf16 P·V)
    err = ((out - ref).abs() / (ref.abs().clamp(min=1.0))).max().item()
    assert err < 2e-2, err
