"""Frame sampling: ``uni_N`` / ``fix_N`` index selection and sliding windows.

Reproduces the reference sampler's *selection semantics* exactly
(reference utils/utils.py:297-333):

- ``"uni_N"``: N uniformly spaced frames;
- ``"fix_N"``: a target rate of N frames/sec → ``round(frame_cnt / fps * N)``
  samples (the reference uses ``int()`` truncation; we keep truncation for
  parity);
- indices come from ``np.linspace(1, frame_cnt - 2, n)`` — the first and last
  frame are deliberately skipped (decoders are flaky at stream edges).

Deliberate divergence: the reference computes milliseconds-per-frame as
``0.001 / fps`` (reference utils/utils.py:312), which is off by 1e6 — the
correct value is ``1000 / fps``.  We emit *correct* millisecond timestamps and
document the difference here; feature shapes are unaffected.
"""
from __future__ import annotations

from typing import List, Tuple

import numpy as np


def parse_extract_method(extract_method: str) -> Tuple[str, float]:
    """Parse ``"uni_12"`` / ``"fix_2"`` into (kind, value)."""
    try:
        kind, val = extract_method.split('_', 1)
        value = float(val)
    except (ValueError, AttributeError) as e:
        raise ValueError(
            f'extract_method must look like "uni_12" or "fix_2", got {extract_method!r}') from e
    if kind not in ('uni', 'fix'):
        raise ValueError(f'extract_method kind must be "uni" or "fix", got {kind!r}')
    if value <= 0:
        raise ValueError(f'extract_method value must be > 0, got {value}')
    return kind, value


def num_samples(extract_method: str, frame_cnt: int, fps: float) -> int:
    """How many frames the method selects from a video of ``frame_cnt`` frames."""
    kind, value = parse_extract_method(extract_method)
    if kind == 'uni':
        # uni_N ALWAYS yields N samples (shape contract (N, C)); indices may
        # repeat for very short videos — same behaviour as the reference's
        # uncapped linspace.
        n = int(value)
    else:  # fix_N: target sampling rate of N fps
        n = int(frame_cnt / fps * value)
    return max(1, n)


def sample_indices(extract_method: str, frame_cnt: int, fps: float) -> np.ndarray:
    """Frame indices selected by the method (int64, ascending, may repeat for
    very short videos — same as the reference's linspace behaviour)."""
    n = num_samples(extract_method, frame_cnt, fps)
    hi = max(frame_cnt - 2, 1)
    # clamp into range: the reference's uncapped linspace indexes frame 1
    # even for a 1-frame video (found by the hypothesis property tests)
    return np.clip(np.linspace(1, hi, n).astype(np.int64), 0,
                   frame_cnt - 1)


def timestamps_ms(indices: np.ndarray, fps: float) -> List[float]:
    """Correct per-frame timestamps in milliseconds (see module docstring for
    the reference's 1e6 bug we deliberately fix)."""
    mspf = 1000.0 / float(fps)
    return [float(i) * mspf for i in indices]


def form_slices(size: int, stack_size: int, step_size: int) -> List[Tuple[int, int]]:
    """Sliding-window ``(start, end)`` pairs over a sequence of ``size`` items.

    Only *full* windows are produced: ``(size - stack) // step + 1`` of them
    (reference utils/utils.py:117-126).
    """
    slices = []
    full_stack_num = (size - stack_size) // step_size + 1
    for i in range(full_stack_num):
        start = i * step_size
        slices.append((start, start + stack_size))
    return slices
