"""Build the sutro_amd._C HIP extension in-tree for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SOURCES = [
    "csrc/bindings.cpp",
    "csrc/rmsnorm.hip",
    "csrc/elementwise.hip",
    "csrc/rope_cache.hip",
    "csrc/qkv_prep.hip",
    "csrc/pool.hip",
    "csrc/attn_decode.hip",
    "csrc/attn_decode_mfma.hip",
    "csrc/attn_prefill.hip",
    "csrc/gemm_tn.hip",
    "csrc/sampler.hip",
    "csrc/grouped_gemm.hip",
]

setup(
    name="sutro-amd-kernels",
    ext_modules=[
        CUDAExtension(
            name="sutro_amd._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
