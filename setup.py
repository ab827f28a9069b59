"""In-tree build of the CDNA4 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built blades_amd/_hip_ops*.so travels with the repo snapshot to GPU
boxes (it is git-ignored but NOT gpurun-ignored).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="blades_amd",
    version="0.1.0",
    packages=["blades_amd", "blades"],
    ext_modules=[
        CUDAExtension(
            name="blades_amd._hip_ops",
            sources=["csrc/blades_kernels.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
