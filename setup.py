"""In-tree build of the gfx950 HIP extensions.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Builds:
  mgwfbp_amd/kernels/mgx_kernels_ext*.so   (fused SGD / pack / unpack / norm)
  mgwfbp_amd/comm/mgx_comm_ext*.so         (RCCL comm core)

The .so files are git-ignored but travel with the gpurun snapshot, so a
CPU-side cross-compile here runs unchanged on the MI355X box.
"""
import os

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

common_args = {
    'cxx': ['-O3', '-std=c++17'],
    'nvcc': ['-O3', '-std=c++17'],
}

ext_modules = [
    CUDAExtension(
        name='mgwfbp_amd.kernels.mgx_kernels_ext',
        sources=['mgwfbp_amd/kernels/mgx_kernels.hip'],
        extra_compile_args=common_args,
    ),
    CUDAExtension(
        name='mgwfbp_amd.kernels.mgx_bn_ext',
        sources=['mgwfbp_amd/kernels/bn_kernels.hip'],
        extra_compile_args=common_args,
    ),
    CUDAExtension(
        name='mgwfbp_amd.comm.mgx_comm_ext',
        sources=['mgwfbp_amd/comm/comm_core.hip'],
        libraries=['rccl'],
        extra_compile_args=common_args,
    ),
]

setup(
    name='mgwfbp_amd',
    version='0.1.0',
    description='MI355X-native merged-gradient WFBP training framework',
    packages=['mgwfbp_amd', 'mgwfbp_amd.comm', 'mgwfbp_amd.kernels',
              'mgwfbp_amd.models', 'mgwfbp_amd.data'],
    ext_modules=ext_modules,
    cmdclass={'build_ext': BuildExtension},
)
