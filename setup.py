"""Build the in-tree HIP/CDNA4 extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in poseidon_amd/ops/ (in-tree, so it travels to GPU boxes
with the repo snapshot)."""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

CSRC = os.path.join("poseidon_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in [
        "bindings.cpp",
        "gemm.hip",
        "elementwise.hip",
        "pool.hip",
        "lrn.hip",
        "softmax.hip",
        "sgd.hip",
        "im2col.hip",
    ]
]

setup(
    name="poseidon_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="poseidon_amd.ops._hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
