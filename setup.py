"""Build the wva_amd native extensions.

Two modules, built in-tree so the .so files travel with the repo snapshot:

- ``wva_amd._queue_native`` — torch binding: CPU (at::parallel_for) path +
  gfx950 HIP batched queue-solver kernel.  Needs torch (and hipcc for the
  GPU object; hipcc cross-compiles on CPU-only machines).
  ``PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace``
- ``wva_amd._queue_native_cpu`` — torch-free pybind11/numpy binding of the
  same host solver (OpenMP).  This is what the slim controller container
  builds: no libtorch, no ROCm.

When torch is not importable only the torch-free module is built.
"""

import os

from setuptools import Extension, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11

ext_modules = []
cmdclass = {}

cpu_ext = Extension(
    "wva_amd._queue_native_cpu",
    sources=["wva_amd/csrc/queue_solver_cpu.cpp"],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-fopenmp", "-std=c++17"],
    extra_link_args=["-fopenmp"],
)
ext_modules.append(cpu_ext)

try:
    from torch.utils.cpp_extension import BuildExtension, CUDAExtension
except ImportError:
    pass
else:
    ext_modules.append(
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
    )
    cmdclass["build_ext"] = BuildExtension

setup(
    name="wva_amd_native",
    ext_modules=ext_modules,
    cmdclass=cmdclass,
)
