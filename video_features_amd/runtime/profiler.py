"""Per-stage wall-clock profiler for extraction jobs (``--profile``).

The reference has no tracing at all; here every extractor
threads its decode / transform / infer / sink work through a
:class:`StageProfiler` so ``--profile`` prints a per-stage breakdown at the
end of each shard.  GPU stages are bracketed with a device synchronize so
the numbers are honest wall-clock, not launch time.
"""
from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict

import torch


class StageProfiler:
    def __init__(self, enabled: bool = False, device: torch.device = None):
        self.enabled = enabled
        self.device = device
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)

    @contextmanager
    def __call__(self, stage: str):
        if not self.enabled:
            yield
            return
        if self.device is not None and self.device.type == 'cuda':
            torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if self.device is not None and self.device.type == 'cuda':
                torch.cuda.synchronize(self.device)
            self.totals[stage] += time.perf_counter() - t0
            self.counts[stage] += 1

    def report(self, header: str = '') -> str:
        if not self.enabled or not self.totals:
            return ''
        total = sum(self.totals.values())
        lines = [f'[profile] {header} (total {total:.3f} s)']
        for stage, t in sorted(self.totals.items(), key=lambda kv: -kv[1]):
            lines.append(f'[profile]   {stage:<12} {t:8.3f} s '
                         f'({100 * t / total:5.1f}%)  x{self.counts[stage]}')
        return '\n'.join(lines)
