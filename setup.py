"""In-tree build of the CDNA4 HIP extension.

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands next to shockwave_amd/ops/ and travels with the repo
snapshot to GPU boxes.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="shockwave_amd_ops",
    ext_modules=[
        CUDAExtension(
            name="shockwave_amd.ops._C",
            sources=[
                "shockwave_amd/ops/csrc/fused_ops.hip",
                "shockwave_amd/ops/csrc/bindings.cpp",
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
