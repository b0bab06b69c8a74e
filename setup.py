"""Build the wva_amd native extension (CPU + gfx950 HIP batched queue solver).

Usage:
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The extension is built in-tree so the .so travels with the repo snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="wva_amd_native",
    ext_modules=[
        CUDAExtension(
            name="wva_amd._queue_native",
            sources=[
                "wva_amd/csrc/queue_solver.cpp",
                "wva_amd/csrc/queue_solver_hip.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-DWVA_WITH_HIP"],
                "nvcc": ["-O3", "-DWVA_WITH_HIP", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
