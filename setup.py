"""In-tree build of the MI355X HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built elasticdl_amd/ops/_C*.so travels with the repo snapshot to GPU
boxes (it is git-ignored but not gpurun-ignored).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="elasticdl_amd.ops._C",
    sources=[
        "elasticdl_amd/ops/csrc/bindings.cpp",
        "elasticdl_amd/ops/csrc/ps_kernels.hip",
        "elasticdl_amd/ops/csrc/train_kernels.hip",
        "elasticdl_amd/ops/csrc/gemm_bf16.hip",
        "elasticdl_amd/ops/csrc/bn_kernels.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="elasticdl_amd",
    version="0.1.0",
    packages=find_packages(include=["elasticdl_amd", "elasticdl_amd.*"]),
    entry_points={
        "console_scripts": ["elasticdl=elasticdl_amd.client.main:main"],
    },
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
