"""Progress reporting shared across worker processes.

The reference shares one tqdm bar across GPU *threads* and closes it
manually to dodge thread races (reference main.py:54-55).  With the
process-per-GPU runtime each rank owns its bar (rank 0 only by default), so
there is no cross-thread mutation to race on.
"""
from __future__ import annotations

import os


class _Null:
    def update(self, n: int = 1) -> None: ...
    def close(self) -> None: ...


def make_progress(total: int, enabled: bool = True):
    if not enabled or os.environ.get('VFA_NO_PROGRESS'):
        return _Null()
    try:
        from tqdm import tqdm
        return tqdm(total=total)
    except Exception:
        return _Null()
