"""bench.py driver-contract tests (CPU): single-rank JSON line and the
2-rank torch.distributed.run path over gloo (mirrors how the driver
launches N>1 GPU runs, with 127.0.0.1 rendezvous)."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json(out: str):
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith('{'):
            return json.loads(line)
    raise AssertionError(f'no JSON line in output:\n{out}')


def test_bench_single_rank_contract():
    r = subprocess.run(
        [sys.executable, 'bench.py', '--steps', '2', '--warmup', '1',
         '--videos-per-step', '2'],
        capture_output=True, text=True, cwd=ROOT, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    d = _last_json(r.stdout)
    for key in ('metric', 'value', 'unit', 'n_gpus', 'steps', 'warmup',
                'ms_per_step', 'higher_is_better', 'scaling', 'vs_baseline',
                'dtype', 'data', 'config'):
        assert key in d, key
    assert d['n_gpus'] == 1 and d['steps'] == 2 and d['warmup'] == 1
    assert d['data'] == 'synthetic' and d['scaling'] == 'weak'
    assert d['value'] > 0
    # the default (flagless) run must measure BASELINE.json's headline
    # metric on its named config
    assert d['metric'] == 'frames/sec CLIP-ViT-B/32 uni_12'
    assert d['config']['model'] == 'CLIP-ViT-B/32'
    assert d['config']['parallelism'] == 'dp1'


def test_bench_two_rank_gloo():
    """The driver's N>1 launch shape: torch.distributed.run, one rank per
    'GPU' (CPU gloo here), rank 0 prints ONE JSON line."""
    env = dict(os.environ)
    env.pop('RANK', None)
    env.pop('WORLD_SIZE', None)
    r = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29517', 'bench.py', '--gpus', '2', '--steps', '2',
         '--warmup', '0', '--videos-per-step', '2'],
        capture_output=True, text=True, cwd=ROOT, env=env, timeout=900)
    assert r.returncode == 0, r.stderr[-3000:]
    d = _last_json(r.stdout)
    assert d['n_gpus'] == 2
    assert d['config']['parallelism'] == 'dp2'
    # whole-job aggregate: 2 ranks x 2 videos x 12 frames x 2 steps
    assert d['value'] > 0
