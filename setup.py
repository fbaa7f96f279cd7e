"""In-tree build of the HIP/CDNA4 extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces dsin_amd/ops/_dsin_hip*.so next to its Python dispatch layer so the
.so travels with the repo snapshot to GPU boxes (no JIT cache dependence).
"""

import glob
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dsin_amd", "ops", "csrc")

sources = sorted(glob.glob(os.path.join(CSRC, "*.hip")) +
                 glob.glob(os.path.join(CSRC, "*.cpp")))

setup(
    name="dsin_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="dsin_amd.ops._dsin_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
