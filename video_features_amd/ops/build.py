"""In-tree hipcc build of the VFA gfx950 extension (no JIT cache — the
built .so lives next to the package so it travels with repo snapshots).

Usage:  python -m video_features_amd.ops.build        (or setup.py build_ext)
"""
from __future__ import annotations

import os
import subprocess
import sysconfig
from pathlib import Path

HERE = Path(__file__).resolve().parent
CSRC = HERE / 'csrc'
OUT = HERE / '_vfa_hip.so'
ARCH = os.environ.get('PYTORCH_ROCM_ARCH', 'gfx950')


def _torch_paths():
    import torch
    troot = Path(torch.__file__).parent
    return troot / 'include', troot / 'include/torch/csrc/api/include', troot / 'lib'


def _newer(src: Path, obj: Path) -> bool:
    return (not obj.exists() or obj.stat().st_mtime < src.stat().st_mtime
            or obj.stat().st_mtime < (CSRC / 'vfa_common.h').stat().st_mtime)


def build(verbose: bool = True) -> Path:
    inc1, inc2, libdir = _torch_paths()
    pyinc = sysconfig.get_paths()['include']
    builddir = HERE / 'build'
    builddir.mkdir(exist_ok=True)
    common = ['-O3', '-std=c++17', '-fPIC', f'--offload-arch={ARCH}',
              '-D__HIP_PLATFORM_AMD__', '-DUSE_ROCM',
              '-DTORCH_EXTENSION_NAME=_vfa_hip']
    objs = []
    relink = not OUT.exists()
    for src in sorted(CSRC.glob('*.hip')) + sorted(CSRC.glob('*.cpp')):
        obj = builddir / (src.stem + '.o')
        objs.append(str(obj))
        if not _newer(src, obj):
            continue
        relink = True
        cmd = ['hipcc', *common, '-c', str(src), '-o', str(obj)]
        if src.suffix == '.cpp':   # bindings need torch + python headers
            cmd += [f'-I{inc1}', f'-I{inc2}', f'-I{pyinc}']
        if verbose:
            print('[vfa build]', ' '.join(cmd), flush=True)
        subprocess.run(cmd, check=True)
    if relink:
        link = ['hipcc', '-shared', '-fPIC', *objs,
                f'-L{libdir}', '-Wl,-rpath,' + str(libdir),
                '-ltorch', '-ltorch_python', '-lc10', '-ltorch_hip',
                '-lc10_hip', '-lamdhip64', '-o', str(OUT)]
        if verbose:
            print('[vfa build]', ' '.join(link), flush=True)
        subprocess.run(link, check=True)
    return OUT


if __name__ == '__main__':
    build()
    print(f'built {OUT}')
