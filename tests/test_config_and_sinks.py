import argparse
import os
import pickle

import numpy as np
import pytest

from video_features_amd.config import Config, sanity_check
from video_features_amd.runtime.sinks import (action_on_extraction,
                                              make_output_path, outputs_exist)


def test_config_from_namespace():
    ns = argparse.Namespace(feature_type='i3d', video_paths=['x.mp4'],
                            flow_type='raft', unknown_flag=42)
    cfg = Config.coerce(ns)
    assert cfg.feature_type == 'i3d'
    assert cfg.flow_type == 'raft'
    assert cfg.on_extraction == 'print'   # default preserved


def test_config_validation():
    with pytest.raises(ValueError):
        Config(feature_type='nope')
    with pytest.raises(ValueError):
        Config(on_extraction='save_hdf5')


def test_sanity_check_rules():
    sanity_check(Config(feature_type='i3d', stack_size=64))
    with pytest.raises(ValueError):
        sanity_check(Config(feature_type='i3d', stack_size=4))
    with pytest.raises(ValueError):
        sanity_check(Config(feature_type='r21d_rgb', extraction_fps=10.0))
    with pytest.raises(ValueError):
        sanity_check(Config(on_extraction='save_numpy', output_path='./x',
                            tmp_path='./x'))
    with pytest.raises(ValueError):
        sanity_check(Config(show_pred=True, device_ids=[0, 1]))


def test_cli_parser_matches_reference_flags():
    from video_features_amd.cli import build_parser
    p = build_parser()
    args = p.parse_args(['--feature_type', 'CLIP-ViT-B/32',
                         '--video_paths', 'a.mp4', 'b.mp4',
                         '--device_ids', '0', '1',
                         '--extract_method', 'uni_12',
                         '--on_extraction', 'save_numpy',
                         '--output_path', '/tmp/out',
                         '--flow_type', 'raft',
                         '--batch_size', '32',
                         '--resize_to_larger_edge',
                         '--side_size', '256'])
    cfg = Config.coerce(args)
    assert cfg.device_ids == [0, 1]
    assert cfg.resize_to_smaller_edge is False
    assert cfg.batch_size == 32


def test_save_numpy_naming(tmp_path):
    feats = {'clip': np.ones((4, 8), np.float32),
             'fps': np.array(25.0),
             'timestamps_ms': np.array([0.0, 40.0])}
    out = str(tmp_path / 'out')
    action_on_extraction(feats, '/data/videos/myvid.mp4', out, 'save_numpy',
                         output_direct=False, feature_type='clip')
    assert sorted(os.listdir(out)) == ['myvid_clip.npy']
    arr = np.load(os.path.join(out, 'myvid_clip.npy'))
    assert arr.shape == (4, 8)
    assert outputs_exist(feats.keys(), '/data/videos/myvid.mp4', out,
                         'save_numpy', False, 'clip')


def test_save_numpy_direct_naming(tmp_path):
    feats = {'clip': np.zeros((2, 2))}
    out = str(tmp_path)
    action_on_extraction(feats, 'v.mp4', out, 'save_numpy',
                         output_direct=True, feature_type='clip')
    assert os.path.exists(os.path.join(out, 'v.npy'))


def test_save_pickle(tmp_path):
    feats = {'i3d_rgb': np.ones((3, 1024)), 'fps': np.array(25.0)}
    out = str(tmp_path)
    action_on_extraction(feats, 'vid.avi', out, 'save_pickle',
                         output_direct=False, feature_type='i3d')
    with open(os.path.join(out, 'vid_i3d.pkl'), 'rb') as f:
        loaded = pickle.load(f)
    assert loaded['i3d_rgb'].shape == (3, 1024)


def test_make_output_path():
    assert make_output_path('/o', 'clip', False) == '/o/clip'
    assert make_output_path('/o', 'clip', True) == '/o'


def test_print_sink_smoke(capsys):
    action_on_extraction({'f': np.arange(6.0).reshape(2, 3)}, 'v.mp4', '.',
                         'print', False, 'f')
    out = capsys.readouterr().out
    assert 'shape (2, 3)' in out


def test_main_cli_end_to_end(tmp_path):
    """python main.py ... on a real file → .npy outputs (the reference's
    primary entry, reference main.py:93-149)."""
    import subprocess
    import sys
    import os
    import numpy as np
    from tests.conftest import synthetic_frames
    from video_features_amd.io.y4m import write_y4m
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, synthetic_frames(t=20, h=64, w=64), fps=25.0)
    r = subprocess.run(
        [sys.executable, 'main.py', '--feature_type', 'resnet18', '--cpu',
         '--video_paths', vid, '--on_extraction', 'save_numpy',
         '--output_path', str(tmp_path / 'out'),
         '--tmp_path', str(tmp_path / 'tmp'), '--batch_size', '8'],
        capture_output=True, text=True, cwd=root, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    out = tmp_path / 'out' / 'resnet18'
    arrs = {f: np.load(out / f) for f in os.listdir(out)}
    feats = [a for f, a in arrs.items() if 'resnet18' in f
             and 'fps' not in f and 'timestamps' not in f]
    assert feats and feats[0].shape == (20, 512)


def test_resume_skips_existing_outputs(tmp_path):
    """--resume: videos whose outputs already exist are skipped (job-level
    resume the reference lacks)."""
    import os
    import time
    import torch
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'v.y4m')
    write_y4m(vid, synthetic_frames(t=10, h=64, w=64), fps=25.0)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[vid], cpu=True,
                 extract_method='uni_2', on_extraction='save_numpy',
                 output_path=str(tmp_path / 'out'),
                 tmp_path=str(tmp_path / 'tmp'), resume=True)
    ExtractCLIP(cfg)(torch.arange(1))
    out_dir = tmp_path / 'out' / 'CLIP-ViT-B/32'
    files = sorted(os.listdir(out_dir))
    assert files
    mtimes = {f: os.path.getmtime(out_dir / f) for f in files}
    time.sleep(0.05)
    ExtractCLIP(cfg)(torch.arange(1))        # resume: must skip
    for f in files:
        assert os.path.getmtime(out_dir / f) == mtimes[f], f
    ExtractCLIP(cfg.replace(resume=False))(torch.arange(1))   # overwrites
    assert any(os.path.getmtime(out_dir / f) != mtimes[f] for f in files)


def test_save_jpg_flow_sink(tmp_path):
    """on_extraction=save_jpg writes per-frame Middlebury flow images
    (works here; the reference's branch is dead, utils/utils.py:96-112)."""
    import os
    import numpy as np
    from video_features_amd.runtime.sinks import action_on_extraction
    flow = np.random.default_rng(0).normal(0, 3, (4, 2, 16, 20)) \
        .astype(np.float32)
    action_on_extraction({'raft': flow, 'fps': np.array(25.0)},
                         '/x/clip.mp4', str(tmp_path), 'save_jpg',
                         False, 'raft')
    files = sorted(os.listdir(tmp_path / 'clip_raft'))
    assert files == [f'flow_{t:06d}.jpg' for t in range(4)]


def test_tmp_transcode_cleanup(tmp_path):
    """tmp_path/{stem}.y4m transcode artifacts are removed per video unless
    --keep_tmp_files."""
    import os
    import torch
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'clip1.mp4.y4m')   # source video elsewhere
    write_y4m(vid, synthetic_frames(t=8, h=64, w=64), fps=25.0)
    tdir = tmp_path / 'tmp'
    tdir.mkdir()
    for keep in (False, True):
        fake = tdir / 'clip1.mp4.y4m'
        write_y4m(str(fake), synthetic_frames(t=8, h=64, w=64), fps=25.0)
        cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[vid],
                     cpu=True, extract_method='uni_2',
                     tmp_path=str(tdir), keep_tmp_files=keep)
        ExtractCLIP(cfg, external_call=True)(torch.arange(1))
        assert os.path.exists(fake) == keep


def test_tmp_cleanup_never_deletes_source(tmp_path):
    import os
    import torch
    from tests.conftest import synthetic_frames
    from video_features_amd.extractors.clip import ExtractCLIP
    from video_features_amd.io.y4m import write_y4m
    vid = str(tmp_path / 'v.y4m')           # source INSIDE tmp_path
    write_y4m(vid, synthetic_frames(t=8, h=64, w=64), fps=25.0)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[vid], cpu=True,
                 extract_method='uni_2', tmp_path=str(tmp_path))
    out = ExtractCLIP(cfg, external_call=True)(torch.arange(1))
    assert len(out) == 1
    assert os.path.exists(vid)


def test_every_feature_type_has_an_extractor():
    from video_features_amd.config import FEATURE_TYPES
    from video_features_amd.models.registry import get_extractor_class
    for ft in FEATURE_TYPES:
        assert get_extractor_class(ft) is not None, ft


def test_cli_args_roundtrip_to_config():
    """Every reference CLI flag parses into the Config API unchanged."""
    from video_features_amd.cli import build_parser
    args = build_parser().parse_args([
        '--feature_type', 'i3d', '--video_paths', 'a.mp4', 'b.mp4',
        '--device_ids', '0', '1', '--flow_type', 'raft',
        '--streams', 'rgb', 'flow', '--stack_size', '24',
        '--step_size', '12', '--extraction_fps', '12.5',
        '--on_extraction', 'save_numpy', '--output_path', '/o',
        '--tmp_path', '/t', '--keep_tmp_files', '--output_direct',
        '--batch_size', '4', '--side_size', '288',
        '--resize_to_larger_edge', '--show_pred', '--temporal_parallel',
        '--gather_features', '--resume', '--profile', '--seed', '7',
        '--dtype', 'bf16'])
    cfg = Config.coerce(args)
    assert cfg.feature_type == 'i3d' and cfg.video_paths == ['a.mp4', 'b.mp4']
    assert cfg.device_ids == [0, 1] and cfg.flow_type == 'raft'
    assert cfg.streams == ['rgb', 'flow']
    assert cfg.stack_size == 24 and cfg.step_size == 12
    assert cfg.extraction_fps == 12.5 and cfg.on_extraction == 'save_numpy'
    assert cfg.keep_tmp_files and cfg.output_direct and cfg.show_pred
    assert cfg.batch_size == 4 and cfg.side_size == 288
    assert cfg.resize_to_smaller_edge is False
    assert cfg.temporal_parallel and cfg.gather_features and cfg.resume
    assert cfg.profile and cfg.seed == 7 and cfg.dtype == 'bf16'


def test_baseline_config1_sample_video():
    """BASELINE.json config 1 verbatim: CLIP-ViT-B/32 --cpu uni_12 on the
    sample clip → (12, 512) features."""
    import os
    import torch
    from video_features_amd.extractors.clip import ExtractCLIP
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    vid = os.path.join(root, 'sample', 'v_GGSY1Qvo990.y4m')
    assert os.path.exists(vid)
    cfg = Config(feature_type='CLIP-ViT-B/32', video_paths=[vid], cpu=True,
                 extract_method='uni_12')
    out = ExtractCLIP(cfg, external_call=True)(torch.arange(1))[0]
    assert out['CLIP-ViT-B/32'].shape == (12, 512)
