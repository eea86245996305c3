"""Build the in-tree HIP extension for gfx950 (MI355X).

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands at deeplearning_amd/ops/_dla_hip*.so and travels with the
repo snapshot to GPU boxes (it is git-ignored, not gpurun-ignored).
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "csrc"

sources = [str(CSRC / "bindings.cpp"), str(CSRC / "cocoeval.cpp"), str(CSRC / "my_add.cpp")] + sorted(
    str(p) for p in CSRC.glob("*.hip") if not p.name.endswith("_hip.hip")
)

setup(
    name="dla_hip",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="deeplearning_amd.ops._dla_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
