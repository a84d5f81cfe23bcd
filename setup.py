"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands at agilerl_amd/ops/_hip_ops*.so (in-tree, so it ships
to GPU boxes with the repo snapshot).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="agilerl_amd.ops._hip_ops",
    sources=[
        "agilerl_amd/ops/csrc/rl_ops.hip",
        "agilerl_amd/ops/csrc/skinny_gemm.hip",
        "agilerl_amd/ops/csrc/lm_ops.hip",
        "agilerl_amd/ops/csrc/norm_ops.hip",
        "agilerl_amd/ops/csrc/act_ops.hip",
        "agilerl_amd/ops/csrc/paged_attn.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="agilerl-amd",
    version="0.1.0",
    packages=find_packages(include=["agilerl_amd", "agilerl_amd.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
