"""In-tree build of the lightctr_amd HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package sources (lightctr_amd/ops/) so the
gpurun snapshot carries it to the GPU box.
"""

import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import glob  # noqa: E402

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                    "lightctr_amd", "ops", "csrc")

# exclude torch-hipify's generated *_hip.* copies (build artifacts)
sources = [os.path.join(CSRC, "bindings.cpp")] + sorted(
    p for p in glob.glob(os.path.join(CSRC, "*.hip"))
    if not p.endswith("_hip.hip"))

ext = CUDAExtension(
    name="lightctr_amd.ops._hip_ops",
    sources=sources,
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="lightctr_amd",
    version="0.1.0",
    packages=["lightctr_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
