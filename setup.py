"""In-tree build of the deepdfa_amd HIP extension for gfx950.

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built deepdfa_amd/_C*.so lives in the package directory so the repo
snapshot that travels to GPU boxes carries it.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="deepdfa_amd._C",
    sources=[
        "csrc/bindings.hip",
        "csrc/flowgnn_kernels.hip",
        "csrc/gemm_bias.hip",
        "csrc/wgrad.hip",
        "csrc/transformer_kernels.hip",
        "csrc/flash_attn.hip",
        "csrc/gemm_bf16.hip",
        "csrc/wgrad2.hip",
        "csrc/adamw.hip",
        "csrc/lmhead_ce.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="deepdfa_amd",
    version="0.1.0",
    packages=["deepdfa_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
