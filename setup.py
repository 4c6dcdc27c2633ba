"""Build the bnsgcn_amd gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces bnsgcn_amd/_C*.so (kept in-tree so the gpurun snapshot carries it).
"""
import os
from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

setup(
    name="bnsgcn_amd",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            "bnsgcn_amd._C",
            sources=["bnsgcn_amd/ops/hip/kernels.hip"],
            extra_compile_args={"nvcc": ["-O3"], "cxx": ["-O3"]},
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
