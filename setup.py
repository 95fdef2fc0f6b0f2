"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting jimm_amd/_hip.*.so is git-ignored but travels with the gpurun
snapshot (source-only history, binary ships to the GPU box).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402 (needs env set first)

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name="jimm_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="jimm_amd._hip",
            sources=[
                "jimm_amd/csrc/bindings.cpp",
                "jimm_amd/csrc/layernorm.hip",
                "jimm_amd/csrc/elementwise.hip",
                "jimm_amd/csrc/adam.hip",
                "jimm_amd/csrc/attention.hip",
                "jimm_amd/csrc/attention_bwd.hip",
                "jimm_amd/csrc/attention_bwd_fused.hip",
                "jimm_amd/csrc/gemm.hip",
                "jimm_amd/csrc/gemm256.hip",
                "jimm_amd/csrc/gemm8p.hip",
                "jimm_amd/csrc/gemm_tn8p.hip",
                "jimm_amd/csrc/losses.hip",
                "jimm_amd/csrc/probe.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
