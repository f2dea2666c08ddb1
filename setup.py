"""In-tree build of the byol_amd HIP extension for gfx950 (MI355X).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(the arch default is set below; hipcc cross-compiles without a GPU).
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "byol_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "ema.hip"),
    os.path.join(CSRC, "byol_loss.hip"),
    os.path.join(CSRC, "lars.hip"),
    os.path.join(CSRC, "bn_fused.hip"),
    os.path.join(CSRC, "augment.hip"),
    os.path.join(CSRC, "cetopk.hip"),
    os.path.join(CSRC, "conv1x1.hip"),
    os.path.join(CSRC, "conv3x3.hip"),
    os.path.join(CSRC, "bn_bf16.hip"),
]

setup(
    name="byol_amd",
    version="0.2.0",
    packages=["byol_amd"],
    ext_modules=[
        CUDAExtension(
            name="byol_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
