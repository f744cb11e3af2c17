"""Build the in-tree HIP extension for the MI355X serving stack.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting production_stack_amd/_C*.so is git-ignored but ships with the
repo snapshot to GPU boxes.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

# CPU-only native gateway pickers (no HIP, no libtorch — the gateway
# sidecar must not pay a torch import): compiled C++ parity with the
# reference's Go endpoint-picker plugins.
import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402

gw_ext = Extension(
    name="production_stack_amd._gwpick",
    sources=["csrc/gateway_pickers.cpp"],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-std=c++17", "-fvisibility=hidden"],
)

ext = CUDAExtension(
    name="production_stack_amd._C",
    sources=[
        "csrc/bindings.cpp",
        "csrc/norm_act_rope.hip",
        "csrc/attention.hip",
        "csrc/prefill_mfma.hip",
        "csrc/prefill_mfma32.hip",
        "csrc/skinny_gemm.hip",
        "csrc/gemm8p.hip",
        "csrc/cachegen.cpp",
        "csrc/lora_bgmv.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": [
            "-O3",
            "-std=c++17",
            "--offload-arch=gfx950",
        ],
    },
)

setup(
    name="production-stack-amd",
    version="0.2.0",
    packages=[
        "production_stack_amd",
        "production_stack_amd.engine",
        "production_stack_amd.engine.models",
        "production_stack_amd.ops",
        "production_stack_amd.router",
        "production_stack_amd.router.examples",
        "production_stack_amd.kvpool",
        "production_stack_amd.parallel",
        "production_stack_amd.gateway",
    ],
    entry_points={
        "console_scripts": [
            "vllm-router=production_stack_amd.router.app:main",
            "ps-engine=production_stack_amd.engine.server:main",
            "ps-kv-controller=production_stack_amd.kvpool.controller:main",
            "ps-cacheserver=production_stack_amd.kvpool.cacheserver:main",
            "ps-endpoint-picker="
            "production_stack_amd.gateway.picker_service:main",
            "ps-extproc=production_stack_amd.gateway.extproc:main",
        ]
    },
    ext_modules=[ext, gw_ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
