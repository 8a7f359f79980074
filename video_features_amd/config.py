"""Extraction configuration.

The reference threads a raw argparse ``Namespace`` through every layer and
external callers fabricate one by hand (reference: README.md:39-51,
main.py:94-137).  That "namespace as API" contract is load-bearing, so this
module keeps it: :class:`Config` is a dataclass constructible from code or from
the CLI with identical semantics, and it *accepts* any object with the same
attribute names (including an ``argparse.Namespace``) via :meth:`Config.coerce`.

Defaults mirror the reference CLI defaults (reference main.py:94-137).
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass
from typing import Any, List, Optional, Sequence

FEATURE_TYPES = [
    'i3d', 'vggish', 'r21d_rgb',
    'resnet18', 'resnet34', 'resnet50', 'resnet101', 'resnet152',
    'raft', 'pwc',
    'CLIP-ViT-B/32', 'CLIP-ViT-B/16', 'CLIP4CLIP-ViT-B-32',
    'CLIP-RN50', 'CLIP-RN101',
    'vggish_torch',
]

ON_EXTRACTION_CHOICES = ['print', 'save_numpy', 'save_pickle', 'save_jpg']
FLOW_TYPES = ['raft', 'pwc', 'flow']


@dataclass
class Config:
    """All knobs of an extraction job.

    Field names intentionally match the reference CLI flags one-to-one so that
    user code written against the reference (``args.feature_type`` etc.) works
    unchanged against a :class:`Config`.
    """

    feature_type: str = 'CLIP-ViT-B/32'

    # ---- inputs (reference utils/utils.py:153-204)
    video_paths: Optional[List[str]] = None
    flow_paths: Optional[List[str]] = None
    file_with_video_paths: Optional[str] = None
    video_dir: Optional[str] = None
    flow_dir: Optional[str] = None

    # ---- devices
    device_ids: Optional[List[int]] = None
    cpu: bool = False

    # ---- tmp / output (reference utils/utils.py:50-114)
    tmp_path: str = './tmp'
    keep_tmp_files: bool = False
    on_extraction: str = 'print'
    output_path: str = './output'
    output_direct: bool = False

    # ---- sampling (reference utils/utils.py:297-333)
    extraction_fps: Optional[float] = None
    extract_method: Optional[str] = None   # "uni_N" or "fix_N"
    stack_size: Optional[int] = None
    step_size: Optional[int] = None

    # ---- streams / flow
    streams: Optional[List[str]] = None     # subset of {'rgb', 'flow'}
    flow_type: str = 'pwc'

    # ---- batching / resize
    batch_size: int = 1
    resize_to_smaller_edge: bool = True
    side_size: Optional[int] = None

    # ---- debug
    show_pred: bool = False

    # ---- new-framework knobs (absent from the reference; additive only)
    dtype: str = 'auto'          # 'auto' | 'fp32' | 'bf16' — compute dtype on GPU
    gather_features: bool = False  # all-gather per-video features to rank 0
    resume: bool = False           # skip videos whose outputs already exist
    profile: bool = False          # per-stage (decode/transform/infer/save) timings
    seed: int = 0                  # random-init weight seed (no-network setting)
    weights_path: Optional[str] = None  # optional state_dict file for the model

    # ---- temporal (context) parallelism: shard ONE video's sliding
    # windows across ranks — exact, windows are independent (the flow
    # "halo" frame is decoded locally); the reference has no equivalent
    temporal_parallel: bool = False
    tp_rank: int = 0      # set by the runtime, not the CLI
    tp_world: int = 1

    def __post_init__(self) -> None:
        if self.feature_type not in FEATURE_TYPES:
            raise ValueError(
                f'unknown feature_type {self.feature_type!r}; choices: {FEATURE_TYPES}')
        if self.on_extraction not in ON_EXTRACTION_CHOICES:
            raise ValueError(
                f'unknown on_extraction {self.on_extraction!r}; choices: {ON_EXTRACTION_CHOICES}')
        if self.flow_type not in FLOW_TYPES:
            raise ValueError(f'unknown flow_type {self.flow_type!r}; choices: {FLOW_TYPES}')

    # ------------------------------------------------------------------
    @classmethod
    def coerce(cls, obj: Any) -> 'Config':
        """Build a Config from any namespace-like object (argparse.Namespace,
        SimpleNamespace, dict, or Config itself). Unknown attributes are
        ignored; missing ones take their defaults."""
        if isinstance(obj, cls):
            return obj
        if isinstance(obj, dict):
            src = obj
        else:
            src = vars(obj)
        names = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in src.items() if k in names})

    def replace(self, **kw: Any) -> 'Config':
        return dataclasses.replace(self, **kw)


def sanity_check(cfg: Config) -> None:
    """Up-front validation, mirroring the reference's rules
    (reference utils/utils.py:129-150):

    - output path must differ from tmp path;
    - ``show_pred`` forces a single device;
    - r21d forbids ``extraction_fps``;
    - i3d requires ``stack_size`` >= 10 when set.
    """
    cfg = Config.coerce(cfg)
    if cfg.on_extraction in ('save_numpy', 'save_pickle'):
        if cfg.output_path == cfg.tmp_path:
            raise ValueError('output_path and tmp_path must differ '
                             f'(both are {cfg.output_path!r})')
    if cfg.show_pred and cfg.device_ids and len(cfg.device_ids) > 1:
        raise ValueError('--show_pred supports a single device only')
    if cfg.temporal_parallel and cfg.feature_type not in ('i3d', 'r21d_rgb'):
        raise ValueError('--temporal_parallel applies to the stack-windowed '
                         "extractors ('i3d', 'r21d_rgb') only")
    if cfg.feature_type == 'r21d_rgb' and cfg.extraction_fps is not None:
        raise ValueError('r21d_rgb does not support custom extraction_fps '
                         '(the pretrained R(2+1)D assumes native fps)')
    if cfg.feature_type == 'i3d' and cfg.stack_size is not None and cfg.stack_size < 10:
        raise ValueError('i3d stack_size must be >= 10')
    if cfg.streams:
        bad = set(cfg.streams) - {'rgb', 'flow'}
        if bad:
            raise ValueError(f'unknown streams {sorted(bad)}; choices: rgb, flow')
    if cfg.dtype not in ('auto', 'fp32', 'bf16'):
        raise ValueError(f"dtype must be 'auto'|'fp32'|'bf16', got {cfg.dtype!r}")
