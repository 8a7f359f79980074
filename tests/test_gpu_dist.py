"""GPU-side distributed-runtime hardening (1 GPU, RCCL).

The 2-rank gloo tests (tests/test_distributed.py) cover the sharding /
gather logic; these cover what gloo CANNOT: RCCL ('nccl' backend on ROCm)
initialization and collectives on a real device, and the exact
``torch.distributed.run … bench.py`` launch path the driver uses — so the
first 8-GPU run is not the code's first execution."""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(('127.0.0.1', 0))
        return s.getsockname()[1]


def test_rccl_one_rank_collectives():
    """init + broadcast + all_gather + all_reduce + barrier over RCCL with
    world_size 1 on cuda:0 — catches RCCL env/IPC issues (e.g. a broken
    HSA_ENABLE_IPC_MODE_LEGACY setup) that gloo tests cannot."""
    import torch.distributed as dist
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', str(_free_port()))
    torch.cuda.set_device(0)
    dist.init_process_group('nccl', rank=0, world_size=1)
    try:
        t = torch.arange(8, device='cuda:0', dtype=torch.float32)
        dist.broadcast(t, src=0)
        out = [torch.empty_like(t)]
        dist.all_gather(out, t)
        assert torch.equal(out[0], t)
        dist.all_reduce(t)
        dist.barrier()
        torch.cuda.synchronize()
    finally:
        dist.destroy_process_group()


def test_rccl_broadcast_models_path():
    """The runtime's broadcast_models() (async broadcast of every param &
    buffer) must run under a live nccl group on device tensors."""
    import torch.distributed as dist
    from video_features_amd.runtime.dist import broadcast_models
    from video_features_amd.models.clip_vit import VisionTransformer
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ['MASTER_PORT'] = str(_free_port())
    torch.cuda.set_device(0)
    dist.init_process_group('nccl', rank=0, world_size=1)
    try:
        m = VisionTransformer().to('cuda:0', torch.bfloat16)
        broadcast_models(m)
        torch.cuda.synchronize()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_torchrun_bench_smoke():
    """One-rank torch.distributed.run of bench.py — the driver's exact
    multi-GPU launch shape (WORLD_SIZE/RANK/LOCAL_RANK from env, rendezvous
    on 127.0.0.1) — must produce the contract JSON line."""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '1', '--master-addr', '127.0.0.1',
           '--master-port', str(_free_port()),
           os.path.join(repo, 'bench.py'), '--gpus', '1',
           '--steps', '2', '--warmup', '1', '--videos-per-step', '16']
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=850,
                       cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines()
            if ln.startswith('{') and '"metric"' in ln]
    assert line, r.stdout[-2000:]
    out = json.loads(line[-1])
    assert out['n_gpus'] == 1 and out['value'] > 0
    assert out['dtype'] == 'bf16'
