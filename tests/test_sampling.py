import numpy as np
import pytest

from video_features_amd.io.sampling import (form_slices, num_samples,
                                            parse_extract_method,
                                            sample_indices, timestamps_ms)


def test_parse_extract_method():
    assert parse_extract_method('uni_12') == ('uni', 12.0)
    assert parse_extract_method('fix_2') == ('fix', 2.0)
    with pytest.raises(ValueError):
        parse_extract_method('bogus')
    with pytest.raises(ValueError):
        parse_extract_method('uni_-1')


def test_uni_sampling_counts_and_edges():
    # reference semantics: linspace(1, frame_cnt-2, N) — skips first/last frame
    idx = sample_indices('uni_12', frame_cnt=100, fps=25.0)
    assert len(idx) == 12
    assert idx[0] == 1 and idx[-1] == 98
    assert (np.diff(idx) >= 0).all()


def test_fix_sampling_counts():
    # fix_2 on a 10s 25fps video (250 frames) → int(250/25*2) = 20 samples
    assert num_samples('fix_2', 250, 25.0) == 20
    idx = sample_indices('fix_2', 250, 25.0)
    assert len(idx) == 20


def test_short_video_does_not_crash():
    idx = sample_indices('uni_12', frame_cnt=3, fps=25.0)
    assert len(idx) == 12       # linspace repeats — same shape contract
    assert idx.max() <= 1


def test_timestamps_correct_ms():
    # the reference's mspf bug (0.001/fps) is deliberately fixed: 1000/fps
    idx = np.array([0, 25, 50])
    ts = timestamps_ms(idx, fps=25.0)
    assert ts == [0.0, 1000.0, 2000.0]


def test_form_slices():
    # reference utils/utils.py:117-126: only full windows
    assert form_slices(10, 4, 2) == [(0, 4), (2, 6), (4, 8), (6, 10)]
    assert form_slices(16, 16, 16) == [(0, 16)]
    assert form_slices(64, 16, 16) == [(0, 16), (16, 32), (32, 48), (48, 64)]
