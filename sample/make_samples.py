#!/usr/bin/env python3
"""Regenerate the deterministic sample clips."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))

from tests.conftest import synthetic_frames
from video_features_amd.io.y4m import write_y4m

here = os.path.dirname(os.path.abspath(__file__))
for name, seed, t in [('v_GGSY1Qvo990', 1, 75), ('v_ZNVhz7ctTq0', 2, 60)]:
    write_y4m(os.path.join(here, f'{name}.y4m'),
              synthetic_frames(t=t, h=128, w=160, seed=seed), fps=25.0)
print('samples written')
