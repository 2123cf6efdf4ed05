"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lives inside the package (glom_pytorch_amd/ops/) so it travels
with repo snapshots to GPU machines; nothing is installed to site-packages.
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ext = CUDAExtension(
    name="glom_pytorch_amd.ops._glom_hip",
    sources=[
        "glom_pytorch_amd/ops/csrc/bindings.cpp",
        "glom_pytorch_amd/ops/csrc/gemm.hip",
        "glom_pytorch_amd/ops/csrc/gemm_fast.hip",
        "glom_pytorch_amd/ops/csrc/aux_kernels.hip",
        "glom_pytorch_amd/ops/csrc/native_ops.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="glom_pytorch_amd",
    version="0.2.0",
    packages=["glom_pytorch_amd", "glom_pytorch_amd.models",
              "glom_pytorch_amd.ops", "glom_pytorch_amd.parallel",
              "glom_pytorch_amd.utils"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
