"""In-tree build of the MI355X-native extensions.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Two extensions:
  xllm_service_amd._ops   HIP/CDNA4 kernels (gfx950) + torch bindings
  (the pure-C++ control-plane core lives in _core; added as it lands)
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
OPS = os.path.join(ROOT, "xllm_service_amd", "csrc", "ops")

ext_modules = [
    CUDAExtension(
        name="xllm_service_amd._ops",
        sources=[
            os.path.join(OPS, f)
            for f in (
                "bindings.cpp",
                "norm.hip",
                "rope.hip",
                "activation.hip",
                "cache.hip",
                "paged_attn_decode.hip",
                "paged_attn_prefill.hip",
                "sampling.hip",
                "gemm.hip",
                "skinny_gemm.hip",
                "packed_gemm.hip",
            )
        ],
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17"],
        },
    ),
]

setup(
    name="xllm_service_amd",
    version="0.1.0",
    packages=["xllm_service_amd"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension},
)
