"""Build the gemscore HIP extension in-tree for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (mpi4dl_amd/_gemscore*.so) so it
travels to GPU boxes with the source snapshot. hipcc cross-compiles on
CPU-only machines.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

here = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="mpi4dl_amd._gemscore",
    sources=[
        "mpi4dl_amd/csrc/gemscore.hip",
        "mpi4dl_amd/csrc/conv_mfma.hip",
        "mpi4dl_amd/csrc/conv_pw.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="mpi4dl_amd",
    version="0.1.0",
    packages=["mpi4dl_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
