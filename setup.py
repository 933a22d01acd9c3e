"""In-tree build of the CDNA4 (gfx950) HIP extension.

`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` produces
dinov3_amd/ops/_hip_ops*.so, which travels with the repo snapshot to GPU
boxes (no JIT cache involved).
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dinov3_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in sorted(os.listdir(CSRC))
    if f.endswith((".hip", ".cpp")) and not f.endswith("_hip.hip")  # skip hipify copies
]

setup(
    name="dinov3_amd_hip_ops",
    ext_modules=[
        CUDAExtension(
            name="dinov3_amd.ops._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
            libraries=["hipblaslt"],
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
