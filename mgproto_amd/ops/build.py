"""Build the in-tree mgproto_hip extension for gfx950.

Usage: ``python -m mgproto_amd.ops.build``. Cross-compiles on CPU-only boxes
(hipcc needs no GPU); the resulting ``_mgproto_hip.so`` sits in-tree so it
travels with repo snapshots.
"""

import os
import shutil
import sys

_HERE = os.path.dirname(os.path.abspath(__file__))
SRC = [os.path.join(_HERE, 'hip', 'mgproto_kernels.hip'),
       os.path.join(_HERE, 'hip', 'fused_bn.hip'),
       os.path.join(_HERE, 'hip', 'em_kernels.hip'),
       os.path.join(_HERE, 'hip', 'enqueue_kernels.hip'),
       os.path.join(_HERE, 'hip', 'gemm1x1_kernels.hip')]
OUT = os.path.join(_HERE, '_mgproto_hip.so')
CPU_SRC = [os.path.join(_HERE, 'cpu', 'fastaug.cpp')]
CPU_OUT = os.path.join(_HERE, '_mgproto_cpu.so')


def build_cpu(verbose: bool = False) -> str:
    """Native CPU augmentation core (plain C++, OpenMP via at::parallel)."""
    from torch.utils.cpp_extension import load
    build_dir = os.path.join(_HERE, 'cpu', 'build')
    os.makedirs(build_dir, exist_ok=True)
    load(name='_mgproto_cpu', sources=CPU_SRC, build_directory=build_dir,
         extra_cflags=['-O3', '-fopenmp'], verbose=verbose,
         is_python_module=True)
    built = os.path.join(build_dir, '_mgproto_cpu.so')
    if os.path.isfile(built):
        shutil.copy2(built, CPU_OUT)
    return CPU_OUT


def build(verbose: bool = False) -> str:
    os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    os.environ.setdefault('MAX_JOBS', '8')
    from torch.utils.cpp_extension import load

    build_dir = os.path.join(_HERE, 'hip', 'build')
    os.makedirs(build_dir, exist_ok=True)
    mod = load(name='_mgproto_hip',
               sources=SRC,
               build_directory=build_dir,
               extra_cflags=['-O3'],
               extra_cuda_cflags=['-O3'],
               verbose=verbose,
               is_python_module=True)
    built = os.path.join(build_dir, '_mgproto_hip.so')
    if os.path.isfile(built):
        shutil.copy2(built, OUT)
    return OUT


if __name__ == '__main__':
    path = build(verbose='-v' in sys.argv)
    print(f'built {path}')
    path = build_cpu(verbose='-v' in sys.argv)
    print(f'built {path}')
