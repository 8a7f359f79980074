#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results DB into per-kernel counter sums.

Usage: python profiles/dump_pmc.py gpurun_out/pmc/*/*_results.db
Schema-defensive: discovers the rocpd table suffix and the pmc/dispatch
join columns at runtime (ROCm 7.2 layouts vary)."""
import glob
import sqlite3
import sys
from collections import defaultdict


def summarize(path: str) -> str:
    db = sqlite3.connect(path)
    cur = db.cursor()
    names = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    try:
        t = next(n for n in names if n.startswith('rocpd_pmc_event'))
    except StopIteration:
        return f'## {path}\n(no pmc_event table; tables: {names})'
    sfx = t[len('rocpd_pmc_event_'):]
    cols = [r[1] for r in cur.execute(f'PRAGMA table_info({t})')]
    out = [f'## {path}', f'(pmc cols: {cols})', '']
    # counter id -> name
    pmc_names = {}
    for n in names:
        if n.startswith('rocpd_info_pmc'):
            try:
                for r in cur.execute(f'SELECT id, name FROM {n}'):
                    pmc_names[r[0]] = r[1]
            except sqlite3.Error:
                pass
    # dispatch -> kernel name
    kd = f'rocpd_kernel_dispatch_{sfx}'
    ks = f'rocpd_info_kernel_symbol_{sfx}'
    disp_kernel = {}
    try:
        for r in cur.execute(
                f'SELECT d.id, s.display_name FROM {kd} d '
                f'JOIN {ks} s ON d.kernel_id = s.id'):
            disp_kernel[r[0]] = r[1][:60]
    except sqlite3.Error as e:
        out.append(f'(dispatch join failed: {e})')
    idc = 'dispatch_id' if 'dispatch_id' in cols else (
        'event_id' if 'event_id' in cols else cols[0])
    pc = 'pmc_id' if 'pmc_id' in cols else ('counter_id' if 'counter_id'
                                            in cols else None)
    vc = 'value' if 'value' in cols else cols[-1]
    sums = defaultdict(float)
    counts = defaultdict(int)
    for r in cur.execute(f'SELECT {idc}, {pc}, {vc} FROM {t}'):
        k = disp_kernel.get(r[0], f'dispatch:{r[0]}')
        c = pmc_names.get(r[1], f'pmc:{r[1]}')
        sums[(k, c)] += r[2]
        counts[(k, c)] += 1
    by_kernel = defaultdict(dict)
    for (k, c), v in sums.items():
        by_kernel[k][c] = v
    for k in sorted(by_kernel):
        out.append(f'### `{k}`')
        row = by_kernel[k]
        for c in sorted(row):
            out.append(f'  {c:<36} {row[c]:.3e}')
        # derived MFMA utilization where the counters allow
        mfma = next((v for c, v in row.items() if 'MFMA_BUSY' in c), None)
        wave = next((v for c, v in row.items() if 'WAVE_CYCLES' in c), None)
        busy = next((v for c, v in row.items() if c.endswith('BUSY_CYCLES')
                     and 'MFMA' not in c), None)
        if mfma and wave:
            # SQ_WAVE_CYCLES counts quad-cycles (MI355X_MICROARCH §PMC)
            out.append(f'  -> MFMA busy / wave cycles ~ '
                       f'{mfma / (4 * wave) * 100:.1f}% (quad-cycle adj)')
        if mfma and busy:
            out.append(f'  -> MFMA busy / SQ busy     ~ '
                       f'{mfma / busy * 100:.1f}%')
        out.append('')
    return '\n'.join(out)


if __name__ == '__main__':
    paths = []
    for a in sys.argv[1:]:
        paths += glob.glob(a)
    for p in paths:
        print(summarize(p))
