"""Cross-implementation numerical parity against the REFERENCE's vendored
nets (pure torch/numpy, CPU).

The round-1 verdict's gap #2: all prior correctness evidence was
self-referential (HIP vs our own torch).  These tests instantiate the
reference's own model sources from /root/reference with random weights, map
the weights into our models (utils/convert_checkpoints + the documented
load paths), and assert feature agreement — a TF-SAME, flatten-order, GRU
merging, or mel-frontend bug diverges here.

Skipped wholesale when /root/reference is unavailable (e.g. on the GPU box;
these are CPU tests and run in the driver's CPU suite).
"""
import os
import sys
import types

import numpy as np
import pytest
import torch

REF = '/root/reference'

pytestmark = pytest.mark.skipif(not os.path.isdir(REF),
                                reason='reference checkout not available')


@pytest.fixture(scope='module')
def ref_path():
    sys.path.insert(0, REF)
    yield REF
    try:
        sys.path.remove(REF)
    except ValueError:
        pass


def _randomize_bn(model, seed=0):
    g = torch.Generator().manual_seed(seed)
    for m in model.modules():
        if isinstance(m, (torch.nn.BatchNorm2d, torch.nn.BatchNorm3d)):
            m.running_mean.copy_(torch.randn(m.running_mean.shape,
                                             generator=g) * 0.3)
            m.running_var.copy_(torch.rand(m.running_var.shape,
                                           generator=g) + 0.5)


# ------------------------------------------------------------------- I3D
@pytest.mark.parametrize('modality', ['rgb', 'flow'])
def test_i3d_feature_parity(ref_path, modality):
    """Our flattened-time I3D must reproduce the reference's conv3d I3D
    (reference models/i3d/i3d_src/i3d_net.py:160-264) bit-for-bit up to
    fp32 accumulation order, weights mapped via convert_auto."""
    from models.i3d.i3d_src.i3d_net import I3D as RefI3D
    from video_features_amd.models.i3d import I3D as OurI3D
    from video_features_amd.utils.convert_checkpoints import convert_auto

    torch.manual_seed(0)
    ref = RefI3D(num_classes=400, modality=modality).eval()
    _randomize_bn(ref, seed=1)
    ours = OurI3D(modality=modality).eval()
    ours.load_state_dict(convert_auto(dict(ref.state_dict())))

    c = 3 if modality == 'rgb' else 2
    g = torch.Generator().manual_seed(2)
    x = torch.rand(1, c, 16, 224, 224, generator=g) * 2 - 1
    with torch.no_grad():
        f_ref = ref(x, features=True)
        f_ours = ours.forward_features(x)
    assert f_ref.shape == f_ours.shape == (1, 1024)
    torch.testing.assert_close(f_ours, f_ref, rtol=1e-4, atol=1e-4)


def test_i3d_logits_parity(ref_path):
    """--show_pred path: the classifier logits must also agree
    (reference i3d_net.py:265-275 softmax head)."""
    from models.i3d.i3d_src.i3d_net import I3D as RefI3D
    from video_features_amd.models.i3d import I3D as OurI3D
    from video_features_amd.utils.convert_checkpoints import convert_auto

    torch.manual_seed(3)
    ref = RefI3D(num_classes=400, modality='rgb').eval()
    _randomize_bn(ref, seed=4)
    ours = OurI3D(modality='rgb').eval()
    ours.load_state_dict(convert_auto(dict(ref.state_dict())))
    g = torch.Generator().manual_seed(5)
    x = torch.rand(1, 3, 16, 224, 224, generator=g) * 2 - 1
    with torch.no_grad():
        # reference forward returns (softmax, logits)
        _, logits_ref = ref(x)
        logits_ours = ours(x)
    torch.testing.assert_close(logits_ours, logits_ref,
                               rtol=1e-4, atol=1e-4)


# ------------------------------------------------------------------- RAFT
def test_raft_flow_parity(ref_path):
    """Our RAFT (merged convzr GRU, persistent buffers, fused-op calls)
    must match the reference RAFT (models/raft/raft_src/raft.py:113-174)
    given its state dict — same iters, same uint8-range inputs."""
    from models.raft.raft_src.raft import RAFT as RefRAFT
    from video_features_amd.models.raft import RAFT as OurRAFT

    torch.manual_seed(6)
    ref = RefRAFT().eval()
    _randomize_bn(ref, seed=7)
    sd = dict(ref.state_dict())
    # the reference aliases each downsample norm twice (norm3 AND
    # downsample.1 are the same module, extractor.py:44-45); keep the
    # downsample.1 naming ours uses
    sd = {k: v for k, v in sd.items() if '.norm3.' not in k}
    ours = OurRAFT().eval()
    missing, unexpected = ours.load_state_dict(sd, strict=False)
    # the SepConvGRU hook merges convz/convr into convzr; nothing else may
    # be missing or unexpected
    assert not missing, missing
    assert not unexpected, unexpected

    # NOTE ≥128 px: below that, the reference's 4-level corr pyramid
    # bottoms out at 1x1 and its bilinear_sampler divides by W-1 = 0
    # (models/raft/raft_src/utils/utils.py:64-65) → NaN.  A reference
    # input-domain limit, not a divergence.
    g = torch.Generator().manual_seed(8)
    im1 = torch.rand(1, 3, 128, 128, generator=g) * 255
    im2 = torch.rand(1, 3, 128, 128, generator=g) * 255
    with torch.no_grad():
        f_ref = ref(im1, im2, iters=6, test_mode=True)
        f_ours = ours(im1, im2, iters=6, test_mode=True)
    assert f_ref.shape == f_ours.shape == (1, 2, 128, 128)
    torch.testing.assert_close(f_ours, f_ref, rtol=1e-3, atol=1e-3)


# ------------------------------------------------------------------ VGGish
def test_vggish_mel_frontend_parity(ref_path):
    """Our torch-native log-mel frontend vs the reference's numpy STFT
    pipeline (models/vggish_torch/vggish_src/mel_features.py:192-223 +
    vggish_input.py example framing, 16 kHz path — no resample)."""
    from models.vggish_torch.vggish_src import mel_features
    from video_features_amd.models.vggish import waveform_to_examples

    rng = np.random.default_rng(9)
    wav = (rng.random(int(16000 * 2.5)) * 2 - 1).astype(np.float64)

    log_mel = mel_features.log_mel_spectrogram(
        wav, audio_sample_rate=16000, log_offset=0.01,
        window_length_secs=0.025, hop_length_secs=0.010,
        num_mel_bins=64, lower_edge_hertz=125, upper_edge_hertz=7500)
    ref_examples = mel_features.frame(log_mel, window_length=96,
                                      hop_length=96)

    ours = waveform_to_examples(torch.from_numpy(wav).float())
    assert ours.shape == ref_examples.shape == (2, 96, 64)
    np.testing.assert_allclose(ours.numpy(), ref_examples,
                               rtol=1e-4, atol=1e-4)


def test_vggish_net_parity(ref_path):
    """Our VGGish net (conv stack + TF-order flatten + FC head) vs the
    reference's VGG (models/vggish_torch/vggish_src/vggish.py:9-31),
    weights mapped via convert_auto."""
    # vggish.py pulls in vggish_input → resampy/soundfile (audio IO not in
    # this image); stub them — VGG itself never touches audio IO
    for name in ('resampy', 'soundfile'):
        sys.modules.setdefault(name, types.ModuleType(name))
    from models.vggish_torch.vggish_src.vggish import VGG, make_layers
    from video_features_amd.models.vggish import VGGish
    from video_features_amd.utils.convert_checkpoints import convert_auto

    torch.manual_seed(10)
    ref = VGG(make_layers()).eval()
    ours = VGGish().eval()
    ours.load_state_dict(convert_auto(dict(ref.state_dict())),
                         strict=False)   # pproc tensors stay ours
    g = torch.Generator().manual_seed(11)
    examples = torch.randn(3, 96, 64, generator=g)
    with torch.no_grad():
        f_ref = ref(examples[:, None, :, :])
        f_ours = ours(examples)
    assert f_ref.shape == f_ours.shape == (3, 128)
    torch.testing.assert_close(f_ours, f_ref, rtol=1e-4, atol=1e-4)


# -------------------------------------------------------------------- PWC
@pytest.fixture()
def ref_pwc_net(ref_path):
    """Import the reference PWC net around its CuPy JIT machinery: stub
    cupy (never called on CPU) and its torch<1 version assert."""
    cupy = types.ModuleType('cupy')
    cupy.util = types.SimpleNamespace(
        memoize=lambda for_each_device=False: (lambda f: f))
    saved_mod = sys.modules.get('cupy')
    sys.modules['cupy'] = cupy
    ver = torch.__version__
    torch.__version__ = '1.6.0'
    try:
        import importlib
        m = importlib.import_module('models.pwc.pwc_src.pwc_net')
        yield m
    finally:
        torch.__version__ = ver
        if saved_mod is not None:
            sys.modules['cupy'] = saved_mod
        else:
            sys.modules.pop('cupy', None)


def test_pwc_net_parity(ref_pwc_net):
    """Our PWC pyramid/warp/decoder vs the reference's vendored net
    (models/pwc/pwc_src/pwc_net.py:212-263) with mapped weights.  The
    reference's correlation is CUDA-only (CuPy); its call site is patched
    to our torch correlation — itself GPU-tested against the HIP kernel —
    so this validates everything around it."""
    from video_features_amd.models.pwc import PWCNet
    from video_features_amd.ops import _pwc_correlation_torch
    from video_features_amd.utils.convert_checkpoints import convert_pwc

    m = ref_pwc_net
    m.correlation.FunctionCorrelation = \
        lambda tensorFirst, tensorSecond, device=None: \
        _pwc_correlation_torch(tensorFirst, tensorSecond, 4)

    torch.manual_seed(12)
    ref = m.PWCNet().eval()
    ours = PWCNet().eval()
    ours.load_state_dict(convert_pwc(dict(ref.state_dict())))
    g = torch.Generator().manual_seed(13)
    im1 = torch.rand(1, 3, 128, 128, generator=g) * 255
    im2 = torch.rand(1, 3, 128, 128, generator=g) * 255
    with torch.no_grad():
        f_ref = ref(im1, im2)
        f_ours = ours(im1, im2)
    assert f_ref.shape == f_ours.shape
    torch.testing.assert_close(f_ours, f_ref, rtol=1e-3, atol=1e-3)
