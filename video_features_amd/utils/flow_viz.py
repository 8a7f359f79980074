"""Optical-flow visualization: Middlebury color wheel → RGB image.

Equivalent of the reference's ``flow_viz.py`` (reference
models/raft/flow_viz.py and models/pwc/flow_viz.py, 131 LoC each —
Baker et al.'s standard color coding, re-implemented vectorized).
"""
from __future__ import annotations

import numpy as np


def make_colorwheel() -> np.ndarray:
    """(55, 3) RGB color wheel: RY/YG/GC/CB/BM/MR arcs of 15/6/4/11/13/6."""
    arcs = [('RY', 15), ('YG', 6), ('GC', 4), ('CB', 11), ('BM', 13), ('MR', 6)]
    total = sum(n for _, n in arcs)
    wheel = np.zeros((total, 3))
    col = 0
    for name, n in arcs:
        t = np.arange(n) / n
        if name == 'RY':
            wheel[col:col + n] = np.stack([np.full(n, 255), 255 * t, np.zeros(n)], 1)
        elif name == 'YG':
            wheel[col:col + n] = np.stack([255 * (1 - t), np.full(n, 255), np.zeros(n)], 1)
        elif name == 'GC':
            wheel[col:col + n] = np.stack([np.zeros(n), np.full(n, 255), 255 * t], 1)
        elif name == 'CB':
            wheel[col:col + n] = np.stack([np.zeros(n), 255 * (1 - t), np.full(n, 255)], 1)
        elif name == 'BM':
            wheel[col:col + n] = np.stack([255 * t, np.zeros(n), np.full(n, 255)], 1)
        else:  # MR
            wheel[col:col + n] = np.stack([np.full(n, 255), np.zeros(n), 255 * (1 - t)], 1)
        col += n
    return wheel


def flow_uv_to_colors(u: np.ndarray, v: np.ndarray) -> np.ndarray:
    wheel = make_colorwheel()
    ncols = wheel.shape[0]
    rad = np.sqrt(u ** 2 + v ** 2)
    angle = np.arctan2(-v, -u) / np.pi
    fk = (angle + 1) / 2 * (ncols - 1)
    k0 = np.floor(fk).astype(np.int32)
    k1 = (k0 + 1) % ncols
    f = (fk - k0)[..., None]
    col = (1 - f) * wheel[k0] / 255.0 + f * wheel[k1] / 255.0
    mask = rad[..., None] <= 1
    col = np.where(mask, 1 - rad[..., None] * (1 - col), col * 0.75)
    return (255 * col).astype(np.uint8)


def flow_to_image(flow: np.ndarray, clip_max: float = None) -> np.ndarray:
    """(H, W, 2) float flow → (H, W, 3) uint8 visualization."""
    u, v = flow[..., 0].copy(), flow[..., 1].copy()
    if clip_max is not None:
        u = np.clip(u, -clip_max, clip_max)
        v = np.clip(v, -clip_max, clip_max)
    rad_max = max(np.sqrt(u ** 2 + v ** 2).max(), 1e-5)
    return flow_uv_to_colors(u / rad_max, v / rad_max)


def save_ppm(path: str, rgb: 'np.ndarray') -> None:
    """Write an (H, W, 3) uint8 image as binary PPM (no PIL dependency)."""
    import numpy as np
    rgb = np.ascontiguousarray(rgb, dtype=np.uint8)
    with open(path, 'wb') as f:
        f.write(b'P6\n%d %d\n255\n' % (rgb.shape[1], rgb.shape[0]))
        f.write(rgb.tobytes())
