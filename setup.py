"""Build the gfx950 HIP extensions IN-TREE:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands inside unionml_amd/ops/ so it travels with the repo
snapshot to GPU machines (no JIT cache dependence).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ext = CUDAExtension(
    name="unionml_amd.ops._tabular_hip",
    sources=[
        "unionml_amd/ops/hip/tabular_ops.cpp",
        "unionml_amd/ops/hip/tabular_kernels.hip",
        "unionml_amd/ops/hip/tabular_gen.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="unionml_amd_ext",
    version="0.1.0",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
