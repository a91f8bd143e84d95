# -*- coding: utf-8 -*-
"""Build the in-tree HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces ``stoke/_C.cpython-*.so`` next to the package so it travels with
repo snapshots (no JIT cache dependence).
"""

import os

from setuptools import setup

from torch.utils import cpp_extension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

setup(
    name="stoke-amd",
    version="0.1.0",
    description="MI355X-native declarative training wrapper for PyTorch-ROCm",
    packages=[
        "stoke",
        "stoke.runtime",
        "stoke.comm",
        "stoke.ddp",
        "stoke.shard",
        "stoke.amp",
        "stoke.ops",
    ],
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="stoke._C",
            sources=["csrc/stoke_kernels.hip", "csrc/fused_bn.hip",
                     "csrc/fused_rmsnorm.hip",
                     "csrc/fused_rope.hip",
                     "csrc/fused_swiglu.hip",
                     "csrc/fused_layernorm.hip",
                     "csrc/fa_fwd.hip",
                     "csrc/fa_bwd.hip",
                     "csrc/fp8_quant.hip",
                     "csrc/fused_ce.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
