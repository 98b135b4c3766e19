"""In-tree build of the gfx950 HIP extension (no hipify, no JIT cache).

Drives hipcc directly: each csrc/*.hip (and bind.cpp) is compiled with
--offload-arch=gfx950 and linked against libtorch into
coinstac_dinunet_amd/ops/_hip_ops.so — the .so lives in the source tree so
it travels with repo snapshots.
"""
import os
import subprocess
import sys
import sysconfig

import torch
from torch.utils import cpp_extension as ce

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, 'csrc')
OUT_SO = os.path.join(HERE, '_hip_ops.so')
ARCH = os.environ.get('PYTORCH_ROCM_ARCH', 'gfx950')


def _newer(paths, target):
    if not os.path.exists(target):
        return True
    t = os.path.getmtime(target)
    return any(os.path.getmtime(p) > t for p in paths)


def build(verbose=True, force=False):
    sources = sorted(
        os.path.join(CSRC, f) for f in os.listdir(CSRC)
        if f.endswith(('.hip', '.cpp')))
    headers = [os.path.join(CSRC, f) for f in os.listdir(CSRC)
               if f.endswith('.h')]
    if not force and not _newer(sources + headers, OUT_SO):
        return OUT_SO

    includes = ce.include_paths() + [sysconfig.get_paths()['include']]
    try:
        import pybind11
        includes.append(pybind11.get_include())
    except ImportError:
        pass
    lib_dir = ce.library_paths()[0]

    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    cflags = ([f'--offload-arch={ARCH}', '-O3', '-std=c++17', '-fPIC',
               '-D__HIP_PLATFORM_AMD__=1', '-DUSE_ROCM=1', '-DHIPBLAS_V2',
               f'-D_GLIBCXX_USE_CXX11_ABI={abi}',
               '-DTORCH_EXTENSION_NAME=_hip_ops',
               '-DTORCH_API_INCLUDE_EXTENSION_H',
               '-Wno-unused-result', '-Wno-deprecated-declarations']
              + [f'-I{d}' for d in includes])

    objs = []
    build_dir = os.path.join(HERE, 'build')
    os.makedirs(build_dir, exist_ok=True)
    for src in sources:
        obj = os.path.join(build_dir,
                           os.path.basename(src).rsplit('.', 1)[0] + '.o')
        if force or _newer([src] + headers, obj):
            cmd = ['hipcc', '-c', src, '-o', obj] + cflags
            if verbose:
                print('[ops.build]', ' '.join(cmd[:4]), '...', flush=True)
            subprocess.run(cmd, check=True)
        objs.append(obj)

    link = (['hipcc', '-shared', '-fPIC', '-o', OUT_SO] + objs +
            [f'-L{lib_dir}', '-ltorch', '-ltorch_cpu', '-ltorch_hip',
             '-lc10', '-lc10_hip', '-ltorch_python',
             f'-Wl,-rpath,{lib_dir}'])
    if verbose:
        print('[ops.build] linking', OUT_SO, flush=True)
    subprocess.run(link, check=True)
    return OUT_SO


if __name__ == '__main__':
    build(force='--force' in sys.argv)
