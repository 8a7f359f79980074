"""BN-folding numerics: folded models must match unfolded eval outputs."""
import pytest
import torch

from video_features_amd.utils.fold_bn import fold_batchnorms


def _randomize_bn_stats(model, seed=0):
    g = torch.Generator().manual_seed(seed)
    for m in model.modules():
        if isinstance(m, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d,
                          torch.nn.BatchNorm3d)):
            m.running_mean.normal_(0, 0.5, generator=g)
            m.running_var.uniform_(0.5, 2.0, generator=g)
            if m.weight is not None:
                m.weight.data.uniform_(0.5, 1.5, generator=g)
                m.bias.data.normal_(0, 0.3, generator=g)


@pytest.mark.parametrize('family', ['resnet50', 'r21d', 'i3d', 'raft_cnet'])
def test_fold_preserves_outputs(family):
    torch.manual_seed(0)
    if family == 'resnet50':
        from video_features_amd.models.resnet import build_resnet
        model = build_resnet('resnet50').eval()
        x = torch.randn(2, 3, 64, 64)
        run = lambda m: m.forward_features(x)
    elif family == 'r21d':
        from video_features_amd.models.r21d import R2Plus1D18
        model = R2Plus1D18().eval()
        x = torch.randn(1, 3, 8, 32, 32)
        run = lambda m: m.forward_features(x)
    elif family == 'i3d':
        from video_features_amd.models.i3d import I3D
        model = I3D(modality='rgb').eval()
        x = torch.randn(1, 3, 16, 64, 64)
        run = lambda m: m.forward_features(x)
    else:
        from video_features_amd.models.raft import BasicEncoder
        model = BasicEncoder(64, 'batch').eval()
        x = torch.randn(2, 3, 64, 64)
        run = lambda m: m(x)
    _randomize_bn_stats(model)
    with torch.no_grad():
        ref = run(model)
        n = fold_batchnorms(model)
        assert n > 0, 'no BNs folded'
        # no BatchNorms left
        for m in model.modules():
            assert not isinstance(m, (torch.nn.BatchNorm2d,
                                      torch.nn.BatchNorm3d))
        out = run(model)
    assert torch.allclose(out, ref, atol=2e-4, rtol=1e-4), \
        (out - ref).abs().max().item()


def test_fold_refuses_training_mode():
    from video_features_amd.models.resnet import build_resnet
    model = build_resnet('resnet18')
    with pytest.raises(AssertionError):
        fold_batchnorms(model)


def test_instance_norm_not_folded():
    from video_features_amd.models.raft import BasicEncoder
    model = BasicEncoder(64, 'instance').eval()
    assert fold_batchnorms(model) == 0
