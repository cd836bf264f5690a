"""Builds the vizier_amd_hip extension (gfx950 HIP kernels) in-tree.

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import os

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

_SRC = [
    'vizier_amd/_src/ops/csrc/module.cpp',
    'vizier_amd/_src/ops/csrc/gram_matern52.hip',
    'vizier_amd/_src/ops/csrc/gram_matern52_bf16.hip',
    'vizier_amd/_src/ops/csrc/gram_matern52_bf16_tiled.hip',
    'vizier_amd/_src/ops/csrc/gram_matern52_fp8.hip',
    'vizier_amd/_src/ops/csrc/gram_matern52_fp8_tiled.hip',
    'vizier_amd/_src/ops/csrc/posterior_score.hip',
    'vizier_amd/_src/ops/csrc/batched_chol.hip',
    'vizier_amd/_src/ops/csrc/eagle_step.hip',
    'vizier_amd/_src/ops/csrc/eagle_sweep.hip',
]

setup(
    name='vizier_amd_hip',
    version='0.1.0',
    ext_modules=[
        CUDAExtension(
            name='vizier_amd_hip',
            sources=_SRC,
            extra_compile_args={
                'cxx': ['-O3'],
                'nvcc': ['-O3', '--offload-arch=gfx950'],
            },
        ),
    ],
    cmdclass={'build_ext': BuildExtension},
)
