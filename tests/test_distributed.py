"""Multi-process runtime tests over gloo (world_size 2, CPU) — the GPU-free
coverage of the RCCL/xGMI path (same code, backend 'nccl' on MI355X)."""
import os

import numpy as np
import pytest
import torch

from video_features_amd.config import Config
from video_features_amd.runtime.dist import (resolve_devices, run_extraction,
                                             shard_indices)


def _make_videos(tmp_path, n=4):
    from tests.conftest import synthetic_frames
    from video_features_amd.io.y4m import write_y4m
    paths = []
    for i in range(n):
        p = str(tmp_path / f'v{i}.y4m')
        write_y4m(p, synthetic_frames(t=8, h=48, w=48, seed=i), fps=25.0)
        paths.append(p)
    return paths


def test_shard_indices_cover_all():
    a = shard_indices(10, 0, 3).tolist()
    b = shard_indices(10, 1, 3).tolist()
    c = shard_indices(10, 2, 3).tolist()
    assert sorted(a + b + c) == list(range(10))
    assert not (set(a) & set(b))


def test_resolve_devices_cpu_multi():
    cfg = Config(cpu=True, device_ids=[0, 1])
    assert resolve_devices(cfg) == ['cpu', 'cpu']
    assert resolve_devices(Config(cpu=True)) == ['cpu']


@pytest.mark.timeout(600)
def test_two_process_gloo_extraction(tmp_path):
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    paths = _make_videos(tmp_path, n=4)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=paths,
                 extract_method='uni_2', cpu=True, device_ids=[0, 1],
                 on_extraction='save_numpy',
                 output_path=str(tmp_path / 'out'),
                 tmp_path=str(tmp_path / 'tmp'))
    run_extraction(cfg)
    out_dir = tmp_path / 'out' / 'CLIP-ViT-B/32'
    files = sorted(os.listdir(out_dir))
    assert len(files) == 4, files   # every shard wrote its videos


@pytest.mark.timeout(600)
def test_two_process_gather_features(tmp_path):
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    paths = _make_videos(tmp_path, n=3)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=paths,
                 extract_method='uni_2', cpu=True, device_ids=[0, 1],
                 gather_features=True, tmp_path=str(tmp_path / 'tmp'))
    feats = run_extraction(cfg)
    assert len(feats) == 3
    for fd in feats:
        assert fd['CLIP-ViT-B/32'].shape == (2, 512)
    # weight broadcast ⇒ both ranks computed with identical weights ⇒ the
    # same video extracted single-process gives identical features
    solo = run_extraction(cfg.replace(device_ids=None, gather_features=True))
    for a, b in zip(feats, solo):
        np.testing.assert_allclose(a['CLIP-ViT-B/32'], b['CLIP-ViT-B/32'],
                                   rtol=1e-4, atol=1e-5)


@pytest.mark.timeout(600)
def test_temporal_parallel_matches_single(tmp_path):
    """--temporal_parallel: one video's sliding windows sharded over 2
    ranks must reproduce the single-process features exactly (windows are
    independent; merge interleaves rank rows back in order)."""
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    from tests.conftest import synthetic_frames
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'long.y4m')
    write_y4m(vid, synthetic_frames(t=45, h=64, w=64), fps=25.0)
    base = Config(feature_type='i3d', video_paths=[vid], cpu=True,
                  stack_size=10, step_size=10, flow_type='pwc',
                  gather_features=True, tmp_path=str(tmp_path / 'tmp'))
    solo = run_extraction(base)
    tp = run_extraction(base.replace(device_ids=[0, 1],
                                     temporal_parallel=True))
    assert len(tp) == len(solo) == 1
    assert tp[0]['rgb'].shape == solo[0]['rgb'].shape == (4, 1024)
    np.testing.assert_allclose(tp[0]['timestamps_ms'],
                               solo[0]['timestamps_ms'])
    np.testing.assert_allclose(tp[0]['rgb'], solo[0]['rgb'],
                               rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(tp[0]['flow'], solo[0]['flow'],
                               rtol=1e-4, atol=1e-4)


@pytest.mark.timeout(600)
def test_temporal_parallel_sink_and_resume(tmp_path):
    """tp mode sinks per video on rank 0 (bounded memory) and honors
    --resume: a second run with existing outputs must not rewrite them."""
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    from tests.conftest import synthetic_frames
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'long.y4m')
    write_y4m(vid, synthetic_frames(t=25, h=64, w=64), fps=25.0)
    base = Config(feature_type='i3d', video_paths=[vid], cpu=True,
                  stack_size=10, step_size=10, flow_type='pwc',
                  on_extraction='save_numpy',
                  output_path=str(tmp_path / 'out'),
                  tmp_path=str(tmp_path / 'tmp'),
                  device_ids=[0, 1], temporal_parallel=True)
    run_extraction(base)
    out_dir = tmp_path / 'out' / 'i3d'
    files = sorted(os.listdir(out_dir))
    assert any('rgb' in f for f in files) and any('flow' in f for f in files)
    mtimes = {f: os.path.getmtime(out_dir / f) for f in files}
    run_extraction(base.replace(resume=True))
    for f, t in mtimes.items():
        assert os.path.getmtime(out_dir / f) == t, f'{f} was rewritten'


def test_temporal_parallel_rejected_for_framewise():
    from video_features_amd.config import sanity_check
    with pytest.raises(ValueError):
        sanity_check(Config(feature_type='resnet50',
                            temporal_parallel=True))


def test_merge_temporal_shards_ragged():
    """Rank 0 may hold one more window than the others; the merge must
    interleave rows back exactly."""
    from video_features_amd.runtime.dist import merge_temporal_shards
    a = {'rgb': np.arange(6).reshape(3, 2), 'fps': np.array(25.0),
         'timestamps_ms': np.array([0.0, 20.0, 40.0])}
    b = {'rgb': np.arange(100, 104).reshape(2, 2), 'fps': np.array(25.0),
         'timestamps_ms': np.array([10.0, 30.0])}
    m = merge_temporal_shards([a, b])
    np.testing.assert_array_equal(
        m['rgb'], [[0, 1], [100, 101], [2, 3], [102, 103], [4, 5]])
    np.testing.assert_array_equal(m['timestamps_ms'],
                                  [0.0, 10.0, 20.0, 30.0, 40.0])
    assert float(m['fps']) == 25.0
