"""Build the theanompi_amd HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces theanompi_amd/_C*.so next to the package sources (the .so ships
to GPU boxes with the repo snapshot).  Kernels are hand-written HIP for
gfx950 only (see theanompi_amd/ops/hip/).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join("theanompi_amd", "ops", "hip")

sources = [
    os.path.join(SRC, "bindings.cpp"),
    os.path.join(SRC, "conv.hip"),
    os.path.join(SRC, "elementwise.hip"),
    os.path.join(SRC, "pool.hip"),
    os.path.join(SRC, "norm.hip"),
    os.path.join(SRC, "softmax.hip"),
]

setup(
    name="theanompi_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="theanompi_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
