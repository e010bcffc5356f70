"""In-tree build of the progen_amd._C HIP extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (progen_amd/_C*.so) so the repo
snapshot carries it to the GPU box.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_SOURCES = [
    "progen_amd/ops/hip/bindings.cpp",
    "progen_amd/ops/hip/ln_shift.hip",
    "progen_amd/ops/hip/glu.hip",
    "progen_amd/ops/hip/cross_entropy.hip",
    "progen_amd/ops/hip/adamw.hip",
    "progen_amd/ops/hip/rope_qkv.hip",
    "progen_amd/ops/hip/attention_fwd.hip",
    "progen_amd/ops/hip/attention_bwd.hip",
    "progen_amd/ops/hip/sgu.hip",
    "progen_amd/ops/hip/fp8_quant.hip",
    "progen_amd/ops/hip/colsum.hip",
]

setup(
    name="progen_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="progen_amd._C",
            sources=HIP_SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
