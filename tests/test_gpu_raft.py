"""GPU numerics tests for the fused RAFT kernels (corr_lookup,
convex_upsample, gru_zr/gru_out) and the full RAFT forward, each against the
plain PyTorch fp32 reference implementation (the ops' own CPU fallbacks).

Run on MI355X with: pytest tests/test_gpu_raft.py -m gpu -x -q
"""
import os

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


def _ref_corr_lookup(pyramid, coords, radius=4):
    """Pure-torch fp32 reference (grid_sample per level, as the reference
    corr.py:36-50 does)."""
    b, _, h, w = coords.shape
    r = radius
    cc = coords.permute(0, 2, 3, 1)
    out = []
    for lvl, corr in enumerate(pyramid):
        # tap t = i*9+j offsets (x + d_i, y + d_j) — the reference's
        # channel order (corr.py:39 adds its (dy,dx)-stacked meshgrid to
        # (x,y) coords); see tests/test_reference_parity.py
        dx = torch.linspace(-r, r, 2 * r + 1, device=coords.device)
        delta = torch.stack(torch.meshgrid(dx, dx, indexing='ij'), dim=-1)
        centroid = cc.reshape(b * h * w, 1, 1, 2) / (2 ** lvl)
        window = (centroid + delta[None])
        gh, gw = corr.shape[-2:]
        gx = window[..., 0] / max(gw - 1, 1) * 2 - 1
        gy = window[..., 1] / max(gh - 1, 1) * 2 - 1
        grid = torch.stack([gx, gy], dim=-1)
        sampled = torch.nn.functional.grid_sample(
            corr, grid, mode='bilinear', padding_mode='zeros',
            align_corners=True)
        out.append(sampled.reshape(b, h, w, -1))
    return torch.cat(out, dim=-1).permute(0, 3, 1, 2).contiguous()


def _pyramid(b, h, w, dev, levels=4, seed=0):
    torch.manual_seed(seed)
    pyr = []
    hh, ww = h, w
    base = torch.randn(b * h * w, 1, h, w, device=dev)
    pyr.append(base.contiguous())
    cur = base
    for _ in range(levels - 1):
        cur = torch.nn.functional.avg_pool2d(cur, 2, 2)
        pyr.append(cur.contiguous())
    return pyr


@pytest.mark.parametrize('nhwc', [False, True])
@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-5),
                                       (torch.bfloat16, 5e-2)])
def test_corr_lookup(dev, nhwc, dtype, tol):
    ops = _hip_loaded()
    b, h, w = 3, 20, 28
    pyr = _pyramid(b, h, w, dev)
    torch.manual_seed(1)
    # coords roaming beyond the borders to exercise zero padding
    coords = (torch.rand(b, 2, h, w, device=dev) * 1.4 - 0.2)
    coords[:, 0] *= w
    coords[:, 1] *= h
    out = ops.corr_lookup(pyr, coords, 4, nhwc, dtype)
    ref = _ref_corr_lookup(pyr, coords)
    assert out.shape == ref.shape == (b, 324, h, w)
    if nhwc:
        assert out.is_contiguous(memory_format=torch.channels_last)
    diff = (out.float() - ref).abs().max().item()
    assert diff < tol, diff


def test_corr_lookup_gmem_path(dev):
    """Feature maps too large for LDS staging take the global-memory
    kernel; numerics must be identical."""
    ops = _hip_loaded()
    b, h, w = 1, 96, 180   # 96*180 + ... > 16384 floats -> gmem kernel
    pyr = _pyramid(b, h, w, dev)
    torch.manual_seed(2)
    coords = torch.rand(b, 2, h, w, device=dev)
    coords[:, 0] *= w
    coords[:, 1] *= h
    out = ops.corr_lookup(pyr, coords, 4, False, torch.float32)
    ref = _ref_corr_lookup(pyr, coords)
    # the torch reference round-trips coords through [-1,1] normalization;
    # at w=180 that costs ~1e-4 of fp32 resolution
    assert (out - ref).abs().max().item() < 5e-4


@pytest.mark.parametrize('nhwc', [False, True])
@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-5),
                                       (torch.bfloat16, 2e-1)])
def test_convex_upsample(dev, nhwc, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    b, h, w = 2, 14, 18
    mf = torch.channels_last if nhwc else torch.contiguous_format
    flow = (torch.randn(b, 2, h, w, device=dev) * 3).to(dtype) \
        .contiguous(memory_format=mf)
    mask = torch.randn(b, 576, h, w, device=dev).to(dtype) \
        .contiguous(memory_format=mf)
    out = ops.convex_upsample(flow, mask, nhwc)
    # fp32 torch reference (the op's own CPU fallback on fp32 inputs)
    m = (0.25 * mask.float()).view(b, 1, 9, 8, 8, h, w).softmax(dim=2)
    up = torch.nn.functional.unfold(8 * flow.float(), 3, padding=1)
    up = (m * up.view(b, 2, 9, 1, 1, h, w)).sum(dim=2)
    ref = up.permute(0, 1, 4, 2, 5, 3).reshape(b, 2, 8 * h, 8 * w)
    assert out.shape == (b, 2, 8 * h, 8 * w)
    assert (out.float() - ref).abs().max().item() < tol


@pytest.mark.parametrize('nhwc', [False, True])
@pytest.mark.parametrize('dtype,tol', [(torch.float32, 1e-6),
                                       (torch.bfloat16, 4e-2)])
def test_gru_gates(dev, nhwc, dtype, tol):
    ops = _hip_loaded()
    torch.manual_seed(0)
    b, c, x, h, w = 2, 16, 24, 10, 12
    mf = torch.channels_last if nhwc else torch.contiguous_format
    zr = torch.randn(b, 2 * c, h, w, device=dev).to(dtype) \
        .contiguous(memory_format=mf)
    q = torch.randn(b, c, h, w, device=dev).to(dtype) \
        .contiguous(memory_format=mf)
    hx = torch.randn(b, c + x, h, w, device=dev).to(dtype) \
        .contiguous(memory_format=mf)
    rhx = torch.randn_like(hx).contiguous(memory_format=mf)
    hx_ref, rhx_ref = hx.clone().float(), rhx.clone().float()

    z = ops.gru_zr(zr, hx, rhx, nhwc)
    ops.gru_out(q, z, hx, nhwc)

    z_ref = torch.sigmoid(zr.float()[:, :c])
    r_ref = torch.sigmoid(zr.float()[:, c:])
    rhx_ref[:, :c] = r_ref * hx_ref[:, :c]
    hx_ref[:, :c] = (1 - z_ref) * hx_ref[:, :c] \
        + z_ref * torch.tanh(q.float())

    assert (z.float() - z_ref).abs().max().item() < tol
    assert (rhx.float() - rhx_ref).abs().max().item() < tol
    assert (hx.float() - hx_ref).abs().max().item() < tol


@pytest.mark.parametrize('nhwc', [False, True])
@pytest.mark.parametrize('iters,min_cos', [(1, 0.98), (4, 0.95)])
def test_raft_full_forward_vs_cpu(dev, nhwc, iters, min_cos):
    """Full RAFT forward on GPU (fused kernels) vs the CPU fp32 reference
    path (same weights, same random inputs).  With random-init weights the
    per-iteration flow deltas are near zero, so fp32 GPU-vs-CPU conv
    rounding is large RELATIVE to the output (measured cos 0.9896 @ 1 iter,
    0.982 @ 4 iters, bit-identical across layouts); the per-op kernels are
    verified at 1e-5..1e-6 in the tests above — this is an integration
    smoke check."""
    from video_features_amd.models.raft import RAFT
    torch.manual_seed(0)
    model = RAFT(iters=iters).eval()
    img1 = torch.randint(0, 256, (2, 3, 64, 96)).float()
    img2 = torch.randint(0, 256, (2, 3, 64, 96)).float()
    with torch.no_grad():
        ref = model(img1, img2, test_mode=True)
        gm = RAFT(iters=iters).eval()
        gm.load_state_dict(model.state_dict())
        gm = gm.to(dev)
        if nhwc:
            gm = gm.use_channels_last()
        out = gm(img1.to(dev), img2.to(dev), test_mode=True).cpu()
    assert out.shape == ref.shape == (2, 2, 64, 96)
    cos = torch.nn.functional.cosine_similarity(
        out.flatten(), ref.flatten(), dim=0).item()
    assert cos > min_cos, cos


def test_raft_bf16_vs_fp32(dev):
    """bf16 GPU RAFT tracks the fp32 GPU result (feature-level agreement)."""
    from video_features_amd.models.raft import RAFT
    torch.manual_seed(0)
    model = RAFT(iters=4).eval().to(dev)
    img1 = torch.randint(0, 256, (2, 3, 64, 96), device=dev).float()
    img2 = torch.randint(0, 256, (2, 3, 64, 96), device=dev).float()
    with torch.no_grad():
        ref = model(img1, img2, test_mode=True).float().cpu()
        bf = model.to(torch.bfloat16).use_channels_last()
        out = bf(img1, img2, test_mode=True).float().cpu()
    cos = torch.nn.functional.cosine_similarity(
        out.flatten(), ref.flatten(), dim=0).item()
    assert cos > 0.9, cos
