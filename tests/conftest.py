import numpy as np
import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: requires a ROCm GPU (run on MI355X with -m gpu)')


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason='no GPU in this environment')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip)


def synthetic_frames(t=16, h=64, w=96, seed=0):
    """Deterministic moving-gradient RGB frames (T, H, W, 3) uint8."""
    rng = np.random.default_rng(seed)
    yy, xx = np.meshgrid(np.arange(h), np.arange(w), indexing='ij')
    frames = np.zeros((t, h, w, 3), np.uint8)
    base = rng.integers(0, 64, size=3)
    for i in range(t):
        frames[i, ..., 0] = (xx * 2 + i * 5 + base[0]) % 256
        frames[i, ..., 1] = (yy * 3 + i * 3 + base[1]) % 256
        frames[i, ..., 2] = ((xx + yy) + i * 7 + base[2]) % 256
    return frames


@pytest.fixture
def frames16():
    return synthetic_frames()


@pytest.fixture
def y4m_video(tmp_path, frames16):
    from video_features_amd.io.y4m import write_y4m
    p = str(tmp_path / 'vid.y4m')
    write_y4m(p, frames16, fps=25.0)
    return p


@pytest.fixture
def npz_video(tmp_path, frames16):
    p = str(tmp_path / 'vid.npz')
    np.savez(p, frames=frames16, fps=25.0)
    return p
