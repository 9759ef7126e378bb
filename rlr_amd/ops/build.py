"""In-tree build of the rlr_amd HIP extension for gfx950.

Drives hipcc directly (no hipify, no JIT cache — the built .so lives
in-tree at rlr_amd/ops/_hip.so so it travels to GPU boxes with the repo
snapshot).  Kernel TUs compile without torch headers (fast); the single
bindings TU includes torch and is linked against libtorch.

Usage: python -m rlr_amd.ops.build [--force]
"""

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, 'csrc')
BUILD = os.path.join(CSRC, 'build')
OUT = os.path.join(HERE, '_hip.so')
ARCH = os.environ.get('PYTORCH_ROCM_ARCH', 'gfx950')

KERNEL_SOURCES = [
    'elementwise.hip', 'flatopt.hip', 'aggregation.hip', 'gemm_f32.hip',
    'gemm_bf16.hip', 'conv_f32.hip', 'conv_bf16.hip', 'conv_bwdw_tap.hip', 'batchnorm.hip',
    'poison.hip',
]
BINDINGS = 'bindings.cpp'


def _torch_paths():
    import torch
    troot = os.path.dirname(torch.__file__)
    return (os.path.join(troot, 'include'),
            os.path.join(troot, 'include', 'torch', 'csrc', 'api', 'include'),
            os.path.join(troot, 'lib'),
            int(torch._C._GLIBCXX_USE_CXX11_ABI))


def _needs_rebuild(src, obj):
    if not os.path.exists(obj):
        return True
    mt = os.path.getmtime(obj)
    if os.path.getmtime(src) > mt:
        return True
    hdr = os.path.join(CSRC, 'common.h')
    return os.path.exists(hdr) and os.path.getmtime(hdr) > mt


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(' '.join(cmd) + '\n' + r.stdout + r.stderr)
        raise RuntimeError(f"build failed: {cmd[-1] if cmd else cmd}")
    return r


def build(force=False, verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    ti, tapi, tlib, abi = _torch_paths()
    py_inc = sysconfig.get_paths()['include']

    objs = []
    jobs = []
    for src in KERNEL_SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.replace('.hip', '.o'))
        objs.append(op)
        if force or _needs_rebuild(sp, op):
            jobs.append((['hipcc', f'--offload-arch={ARCH}', '-O3',
                          '-std=c++17', '-fPIC', '-c', sp, '-o', op], src))

    bsrc = os.path.join(CSRC, BINDINGS)
    bobj = os.path.join(BUILD, 'bindings.o')
    objs.append(bobj)
    if force or _needs_rebuild(bsrc, bobj):
        jobs.append((['hipcc', f'--offload-arch={ARCH}', '-O2', '-std=c++17',
                      '-fPIC', '-DTORCH_EXTENSION_NAME=_hip',
                      f'-D_GLIBCXX_USE_CXX11_ABI={abi}',
                      f'-I{ti}', f'-I{tapi}', f'-I{py_inc}',
                      '-c', bsrc, '-o', bobj], BINDINGS))

    if jobs:
        if verbose:
            print(f"[rlr_amd.ops.build] compiling {len(jobs)} TU(s) "
                  f"for {ARCH}")
        with ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(lambda j: _run(j[0]), jobs))

    if jobs or force or not os.path.exists(OUT) or any(
            os.path.getmtime(o) > os.path.getmtime(OUT) for o in objs):
        link = ['hipcc', '-shared', '-fPIC', *objs, '-o', OUT,
                f'-L{tlib}', '-ltorch', '-ltorch_cpu', '-ltorch_python',
                '-lc10', '-lc10_hip', '-ltorch_hip', '-lamdhip64',
                f'-Wl,-rpath,{tlib}']
        if verbose:
            print("[rlr_amd.ops.build] linking _hip.so")
        _run(link)
    if verbose:
        print(f"[rlr_amd.ops.build] ok: {OUT}")
    return OUT


if __name__ == '__main__':
    build(force='--force' in sys.argv)
