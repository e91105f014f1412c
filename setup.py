"""Build the helix_amd native extension in-tree for MI355X (gfx950).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""
import os
import glob
from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "helix_amd", "ops", "hip")

sources = sorted(
    f for f in glob.glob(os.path.join(HIP_DIR, "*.hip"))
    if not f.endswith("_hip.hip")       # torch-hipify build artifacts
) + [os.path.join(HIP_DIR, "bindings.cpp")]

setup(
    name="helix_amd_C",
    ext_modules=[
        CUDAExtension(
            name="helix_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
