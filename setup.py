# -*- coding: utf-8 -*-
"""In-tree build of the fedtorch_amd CDNA4 kernel pack.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces fedtorch_amd/ops/_C*.so next to the Python wrappers so the built
extension travels with the repo snapshot (gpurun / judge environments).
"""
import os

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name='fedtorch_amd_ops',
    ext_modules=[
        CUDAExtension(
            name='fedtorch_amd.ops._C',
            sources=['fedtorch_amd/ops/hip/ops.hip'],
            extra_compile_args={
                'cxx': ['-O3'],
                'nvcc': ['-O3', '--offload-arch=gfx950'],
            },
        ),
    ],
    cmdclass={'build_ext': BuildExtension.with_options(no_python_abi_suffix=False)},
)
