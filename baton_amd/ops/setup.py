"""In-tree build of the baton_amd HIP extension for gfx950.

Run from this directory (or via python -m baton_amd.ops.build):
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting _hip_ops*.so sits next to the Python wrappers so the repo
snapshot carries it to GPU boxes (no JIT cache dependence).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = [
    os.path.join(HERE, "csrc", f)
    for f in [
        "bindings.cpp",
        "optim.hip",
        "fedmath.hip",
        "loss.hip",
        "norm.hip",
        "elementwise.hip",
        "gemm.hip",
        "gemm8.hip",
        "attention.hip",
        "flash.hip",
        "llama_ops.hip",
        "conv.hip",
        "conv8.hip",
        "conv_halo.hip",
    ]
]

setup(
    name="baton_amd_hip_ops",
    ext_modules=[
        CUDAExtension(
            "_hip_ops",
            SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
