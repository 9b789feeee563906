"""Build the roc_amd native extension (roc_amd._C) in-tree for gfx950.

Usage:  python setup.py build_ext --inplace

All device code is hand-written HIP for CDNA4 (gfx950) — no hipify output,
no CUDA compatibility paths. The extension is built in-tree so the .so
travels with the repo snapshot to GPU boxes.
"""
import os
import glob

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import CUDAExtension, BuildExtension


def sources():
    srcs = sorted(glob.glob("roc_amd/csrc/*.cpp"))
    # exclude torch-hipify's generated *_hip.hip copies from earlier
    # builds (gitignored, but present in a built tree)
    srcs += sorted(s for s in glob.glob("roc_amd/csrc/*.hip")
                   if not s.endswith("_hip.hip"))
    return srcs


setup(
    name="roc_amd",
    version="0.1.0",
    description="MI355X-native distributed full-graph GNN training framework",
    packages=["roc_amd"],
    ext_modules=[
        CUDAExtension(
            name="roc_amd._C",
            sources=sources(),
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17", "-fopenmp"],
                "nvcc": [
                    "-O3",
                    "-std=c++17",
                    "--offload-arch=gfx950",
                ],
            },
            extra_link_args=["-fopenmp"],
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
