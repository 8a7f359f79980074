"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference.

Run on MI355X with: pytest tests/test_gpu_ops.py -m gpu -x -q
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def _hip_loaded():
    from video_features_amd import ops
    assert ops.hip_available(), 'HIP extension must be built in-tree'
    return ops


def test_extension_loaded_native(dev):
    ops = _hip_loaded()
    from video_features_amd.ops import _vfa_hip
    assert _vfa_hip.gfx_arch == 'gfx950'
    assert '_vfa_hip' in _vfa_hip.__file__


@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-5),
                                       (torch.bfloat16, 2e-2)])
def test_quick_gelu(dev, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = torch.randn(1000003, device=dev, dtype=dtype)
    out = ops.quick_gelu(x).float()
    ref = (x.float() * torch.sigmoid(1.702 * x.float()))
    assert (out - ref).abs().max().item() < tol


@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-5),
                                       (torch.bfloat16, 2e-2)])
def test_gelu_tanh(dev, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = torch.randn(65537, device=dev, dtype=dtype)
    out = ops.gelu(x).float()
    ref = torch.nn.functional.gelu(x.float(), approximate='tanh')
    assert (out - ref).abs().max().item() < tol


@pytest.mark.parametrize('d', [768, 1024, 100])
@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-4),
                                       (torch.bfloat16, 6e-2)])
def test_layer_norm(dev, d, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = torch.randn(517, d, device=dev, dtype=dtype)
    w = torch.randn(d, device=dev, dtype=dtype)
    b = torch.randn(d, device=dev, dtype=dtype)
    out = ops.layer_norm(x, w, b).float()
    ref = torch.nn.functional.layer_norm(x.float(), (d,), w.float(), b.float())
    assert (out - ref).abs().max().item() < tol


@pytest.mark.parametrize('n,d', [(50, 64), (64, 64), (49, 32), (10, 128)])
def test_mhsa_vs_fp32(dev, n, d):
    ops = _hip_loaded()
    torch.manual_seed(0)
    q, k, v = [torch.randn(4, 6, n, d, device=dev, dtype=torch.bfloat16)
               for _ in range(3)]
    out = ops.attention(q, k, v).float()
    scale = 1.0 / (d ** 0.5)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), scale=scale)
    err = (out - ref).abs().max().item()
    assert err < 3e-2, err


@pytest.mark.parametrize('c', [32, 64, 96, 128, 196])
@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-4),
                                       (torch.bfloat16, 3e-2)])
def test_pwc_correlation(dev, c, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    f1 = torch.randn(2, c, 30, 37, device=dev, dtype=dtype)
    f2 = torch.randn(2, c, 30, 37, device=dev, dtype=dtype)
    out = ops.pwc_correlation(f1, f2).float()
    ref = ops._pwc_correlation_torch(f1.float(), f2.float(), 4)
    err = (out - ref).abs().max().item()
    assert err < tol, (c, err)


@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-4),
                                       (torch.bfloat16, 4e-2)])
def test_bilinear_warp(dev, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = torch.randn(2, 16, 24, 31, device=dev, dtype=dtype)
    flow = torch.randn(2, 2, 24, 31, device=dev, dtype=dtype) * 3
    out = ops.bilinear_warp(x, flow).float()
    import os
    os.environ['VFA_FORCE_TORCH_OPS'] = '1'
    try:
        ref = ops.bilinear_warp(x.float(), flow.float())
    finally:
        del os.environ['VFA_FORCE_TORCH_OPS']
    err = (out - ref).abs().max().item()
    assert err < tol, err


@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-4),
                                       (torch.bfloat16, 4e-2)])
def test_grid_sample(dev, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    x = torch.randn(64, 1, 24, 31, device=dev, dtype=dtype)
    coords = (torch.rand(64, 9, 9, 2, device=dev, dtype=dtype) * 36 - 3)
    out = ops.grid_sample_bilinear(x, coords).float()
    import os
    os.environ['VFA_FORCE_TORCH_OPS'] = '1'
    try:
        ref = ops.grid_sample_bilinear(x.float(), coords.float())
    finally:
        del os.environ['VFA_FORCE_TORCH_OPS']
    err = (out - ref).abs().max().item()
    assert err < tol, err


def test_clip_vit_gpu_forward(dev):
    """Full flagship model forward on GPU in bf16 through the HIP ops."""
    from video_features_amd.models.clip_vit import VisionTransformer
    torch.manual_seed(0)
    m = VisionTransformer().to(dev, torch.bfloat16).eval()
    x = torch.randn(12, 3, 224, 224, device=dev, dtype=torch.bfloat16)
    with torch.no_grad():
        out = m.encode_image(x)
    assert out.shape == (12, 512)
    assert torch.isfinite(out.float()).all()


def test_clip_vit_gpu_matches_cpu_fp32(dev):
    """The bf16 GPU pipeline must agree with the CPU fp32 reference to bf16
    tolerance (cosine similarity of features)."""
    from video_features_amd.models.clip_vit import ViTConfig, VisionTransformer
    torch.manual_seed(0)
    cfg = ViTConfig(input_resolution=64, patch_size=16, width=128, layers=4,
                    heads=4, output_dim=64)
    m = VisionTransformer(cfg).eval()
    x = torch.randn(8, 3, 64, 64)
    with torch.no_grad():
        ref = m.encode_image(x)
        mg = m.to(dev, torch.bfloat16)
        out = mg.encode_image(x.to(dev, torch.bfloat16)).float().cpu()
    cos = torch.nn.functional.cosine_similarity(out, ref).min().item()
    assert cos > 0.99, cos


def test_clip_vit_b16_gpu_matches_cpu_fp32(dev):
    """ViT-B/16 = 197 tokens: covers the multi-KV-tile flash path in a
    full-model forward."""
    from video_features_amd.models.clip_vit import build_clip_vit
    torch.manual_seed(0)
    m = build_clip_vit('CLIP-ViT-B/16').eval()
    x = torch.randn(2, 3, 224, 224)
    with torch.no_grad():
        ref = m.encode_image(x)
        out = m.to(dev).encode_image(x.to(dev)).cpu()
    cos = torch.nn.functional.cosine_similarity(out.flatten(), ref.flatten(),
                                                dim=0).item()
    assert cos > 0.999, cos
