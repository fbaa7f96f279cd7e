"""In-tree build of the HIP/CDNA4 extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces dsin_amd/ops/_dsin_hip*.so next to its Python dispatch layer. The
.so is git-ignored (history stays source-only) but does travel with gpurun
snapshots to GPU boxes, so there is no JIT-cache dependence at run time.
"""

import glob
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dsin_amd", "ops", "csrc")

# Exclude hipify-generated "*_hip.hip" twins: torch's hipify pass recreates
# them at build time from the hand-written sources; globbing both would
# compile duplicate symbols (and the twins can drift from the originals).
sources = sorted(s for s in
                 glob.glob(os.path.join(CSRC, "*.hip")) +
                 glob.glob(os.path.join(CSRC, "*.cpp"))
                 if not s.endswith("_hip.hip"))

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
