"""Builds the in-tree gfx950 HIP extension:

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting lingvo_amd/ops/_lingvo_ops*.so is git-ignored but travels to
GPU boxes with the repo snapshot (see __graft_entry__.build()).
"""

import glob
import os

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils import cpp_extension

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, 'lingvo_amd', 'ops', 'hip')

sources = sorted(
    glob.glob(os.path.join(HIP_DIR, '*.hip')) +
    glob.glob(os.path.join(HIP_DIR, '*.cpp')))

setup(
    name='lingvo_amd_ops',
    ext_modules=[
        cpp_extension.CUDAExtension(
            name='lingvo_amd.ops._lingvo_ops',
            sources=sources,
            include_dirs=[HIP_DIR],
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17', '--offload-arch=gfx950'],
            },
        )
    ],
    cmdclass={'build_ext': cpp_extension.BuildExtension},
)
