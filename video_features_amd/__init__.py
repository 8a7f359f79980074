"""video_features_amd — MI355X-native video feature extraction framework.

A from-scratch rebuild of the capabilities of Kamino666/video_features
(CLIP / I3D / R(2+1)D / ResNet / RAFT / PWC / VGGish feature extraction from
videos, fanned out data-parallel across GPUs), designed for AMD Instinct
MI355X: PyTorch-ROCm orchestration, hand-written CDNA4 HIP kernels for the
hot ops, and RCCL over xGMI for multi-GPU runs.

Library API (mirrors the reference's external-call contract,
reference README.md:38-57)::

    from video_features_amd import Config, ExtractCLIP
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=['a.y4m'],
                 extract_method='uni_12')
    extractor = ExtractCLIP(cfg, external_call=True)
    feats_list = extractor(torch.arange(1))
    feats_list[0]['CLIP-ViT-B/32']   # (12, 512) np.ndarray
"""
from .config import Config, sanity_check, FEATURE_TYPES
from .models.registry import get_extractor_class


def __getattr__(name):
    # lazy extractor exports: ExtractCLIP, ExtractResNet, ...
    lazy = {
        'ExtractCLIP': ('extractors.clip', 'ExtractCLIP'),
        'ExtractResNet': ('extractors.resnet', 'ExtractResNet'),
        'ExtractR21D': ('extractors.r21d', 'ExtractR21D'),
        'ExtractI3D': ('extractors.i3d', 'ExtractI3D'),
        'ExtractRAFT': ('extractors.raft', 'ExtractRAFT'),
        'ExtractPWC': ('extractors.pwc', 'ExtractPWC'),
        'ExtractVGGish': ('extractors.vggish', 'ExtractVGGish'),
        'run_extraction': ('runtime.dist', 'run_extraction'),
    }
    if name in lazy:
        import importlib
        mod, attr = lazy[name]
        return getattr(importlib.import_module(f'.{mod}', __name__), attr)
    raise AttributeError(f'module {__name__!r} has no attribute {name!r}')


__version__ = '0.1.0'
__all__ = ['Config', 'sanity_check', 'FEATURE_TYPES', 'get_extractor_class',
           'ExtractCLIP', 'ExtractResNet', 'ExtractR21D', 'ExtractI3D',
           'ExtractRAFT', 'ExtractPWC', 'ExtractVGGish', 'run_extraction']
