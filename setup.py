"""Build the sagecal_amd HIP extension in-tree for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in sagecal_amd/ops/hip/ so it travels with the repo snapshot
to GPU boxes (no JIT cache dependence).
"""
import os
import sys

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(ROOT, 'sagecal_amd', 'ops', 'hip')

ext = CUDAExtension(
    name='sagecal_amd.ops.hip.dirac_hip',
    sources=[
        os.path.join(SRC, 'bindings.cpp'),
        os.path.join(SRC, 'launchers.hip'),
    ],
    extra_compile_args={
        'cxx': ['-O3', '-std=c++17'],
        'nvcc': ['-O3', '-std=c++17', '--offload-arch=gfx950'],
    },
)

setup(
    name='sagecal_amd',
    version='0.1.0',
    packages=['sagecal_amd', 'sagecal_amd.ops', 'sagecal_amd.ops.hip',
              'sagecal_amd.solvers', 'sagecal_amd.consensus',
              'sagecal_amd.apps', 'sagecal_amd.utils'],
    ext_modules=[ext],
    cmdclass={'build_ext': BuildExtension.with_options(no_python_abi_suffix=True)},
)
