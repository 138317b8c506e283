"""Build the in-tree gfx950 HIP kernel extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands next to the package (gllm_amd/_kernels*.so) so it travels
with repo snapshots (gpurun) without a site-packages install.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

CSRC = os.path.join("gllm_amd", "ops", "csrc")

ext = CUDAExtension(
    name="gllm_amd._kernels",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "norm.hip"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "attention_decode.hip"),
        os.path.join(CSRC, "attention_prefill.hip"),
        os.path.join(CSRC, "attention_mla.hip"),
        os.path.join(CSRC, "moe.hip"),
        os.path.join(CSRC, "fp8.hip"),
        os.path.join(CSRC, "gdn.hip"),
        os.path.join(CSRC, "int4.hip"),
        os.path.join(CSRC, "skinny_gemm.hip"),
        os.path.join(CSRC, "sampling.hip"),
        os.path.join(CSRC, "custom_ar.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="gllm_amd",
    version="0.1.0",
    packages=["gllm_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
