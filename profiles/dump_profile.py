#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB (kernel-trace) into a markdown table.

Usage: python profiles/dump_profile.py gpurun_out/prof2/runc/*_results.db > profiles/xxx.md
"""
import glob
import sqlite3
import sys


def summarize(path: str, tail: float = 0.0) -> str:
    db = sqlite3.connect(path)
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table' "
        "AND name LIKE 'rocpd_kernel_dispatch%'")][0]
    sfx = t[len('rocpd_kernel_dispatch_'):]
    where = ''
    if tail > 0:
        lo, hi = cur.execute(
            f'SELECT MIN(start), MAX(end) FROM rocpd_kernel_dispatch_{sfx}'
        ).fetchone()
        cut = hi - (hi - lo) * tail
        where = f' WHERE k.start >= {cut} '
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*) n, SUM(k.end-k.start)/1e6 ms,
               AVG(k.end-k.start)/1e3 avg,
               MAX(ks.arch_vgpr_count), MAX(ks.sgpr_count), MAX(ks.group_segment_size)
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
        {where}
        GROUP BY ks.display_name ORDER BY ms DESC LIMIT 25""").fetchall()
    tot = cur.execute(
        f"SELECT SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{sfx} k"
        + where).fetchone()[0]
    out = [f'## {path}', '',
           '| total ms | % | calls | avg us | vgpr | sgpr | lds B | kernel |',
           '|---|---|---|---|---|---|---|---|']
    for name, n, ms, avg, vgpr, sgpr, lds in rows:
        out.append(f'| {ms:.2f} | {100*ms/tot:.1f} | {n} | {avg:.1f} | '
                   f'{vgpr} | {sgpr} | {lds} | `{name[:80]}` |')
    out.append(f'\nGPU kernel time total: {tot:.2f} ms')
    return '\n'.join(out)


if __name__ == '__main__':
    tail = 0.0
    args = []
    for a in sys.argv[1:]:
        if a.startswith('--tail='):
            tail = float(a.split('=')[1])
        else:
            args.append(a)
    paths = []
    for a in args:
        paths += glob.glob(a)
    for p in paths:
        print(summarize(p, tail))
        print()
