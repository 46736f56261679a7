#!/usr/bin/env python3
"""In-tree build of the distar_amd HIP/CDNA4 extension (_hip_ops.so).

Drives hipcc directly (`hipcc --offload-arch=gfx950`) so the .so lands in
distar_amd/ops/ and travels to GPU boxes with the repo snapshot (no JIT
cache dependency).  Usage: `python setup_hip.py build`.
"""
import os
import subprocess
import sys
import sysconfig

import torch
import torch.utils.cpp_extension as cpp_ext

ROOT = os.path.dirname(os.path.abspath(__file__))
OPS = os.path.join(ROOT, 'distar_amd', 'ops')
HIP_DIR = os.path.join(OPS, 'hip')
OUT = os.path.join(OPS, '_hip_ops.so')
ARCH = os.environ.get('PYTORCH_ROCM_ARCH', 'gfx950')

SOURCES = [
    os.path.join(HIP_DIR, 'scans.hip'),
    os.path.join(HIP_DIR, 'lnlstm.hip'),
    os.path.join(HIP_DIR, 'upsample.hip'),
    os.path.join(HIP_DIR, 'entity_embed.hip'),
    os.path.join(HIP_DIR, 'su_sample.hip'),
    os.path.join(HIP_DIR, 'ce_loss.hip'),
    os.path.join(HIP_DIR, 'rl_rowwise.hip'),
    os.path.join(HIP_DIR, 'entity_attn.hip'),
    os.path.join(HIP_DIR, 'scatter.hip'),
    os.path.join(HIP_DIR, 'residual_ln.hip'),
    os.path.join(HIP_DIR, 'conv2d.hip'),
    os.path.join(HIP_DIR, 'multi_tensor.hip'),
    os.path.join(HIP_DIR, 'bindings.cpp'),
]


def build_native_codec(verbose=True):
    """CPU-only pybind codec (no hipcc needed): g++ + zlib."""
    import pybind11
    src = os.path.join(OPS, 'cpp', 'native_codec.cpp')
    out = os.path.join(OPS, '_native_codec.so')
    py_inc = sysconfig.get_paths()['include']
    cmd = ['g++', '-O3', '-std=c++17', '-shared', '-fPIC', src, '-o', out,
           f'-I{pybind11.get_include()}', f'-I{py_inc}', '-lz']
    if verbose:
        print(' '.join(cmd))
    subprocess.check_call(cmd, cwd=ROOT)
    return out


def build(verbose=True):
    build_native_codec(verbose=verbose)
    sources = [s for s in SOURCES if os.path.exists(s)]
    torch_inc = cpp_ext.include_paths()
    torch_lib = cpp_ext.library_paths()
    py_inc = sysconfig.get_paths()['include']
    abi = int(torch.compiled_with_cxx11_abi())
    cmd = [
        'hipcc', f'--offload-arch={ARCH}', '-O3', '-std=c++17', '-fPIC',
        '-shared', '-o', OUT,
        f'-D_GLIBCXX_USE_CXX11_ABI={abi}',
        '-DTORCH_EXTENSION_NAME=_hip_ops',
        '-DTORCH_API_INCLUDE_EXTENSION_H',
        '-DUSE_ROCM', '-D__HIP_PLATFORM_AMD__=1',
        '-fno-gpu-rdc',
        '-Wno-unused-result',
    ]
    for inc in torch_inc + [py_inc]:
        cmd.append(f'-I{inc}')
    cmd += sources
    for lib in torch_lib:
        cmd.append(f'-L{lib}')
        cmd.append(f'-Wl,-rpath,{lib}')
    cmd += ['-ltorch', '-ltorch_python', '-ltorch_hip', '-lc10', '-lc10_hip',
            '-lamdhip64']
    if verbose:
        print(' '.join(cmd))
    subprocess.check_call(cmd, cwd=ROOT)
    print(f'built {OUT}')
    return OUT


if __name__ == '__main__':
    if len(sys.argv) > 1 and sys.argv[1] not in ('build', 'build_ext'):
        print(__doc__)
        sys.exit(1)
    build()
