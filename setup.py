"""In-tree build of the lfmq_hip extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built lfmq_hip*.so lands at the repo root so it travels with gpurun
snapshots (a JIT cache under ~/.cache would not).
"""
import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

# LFMQ_ASAN=1: address-sanitized build of the HOST side of the extension
# (binding/launch logic).  Run with
# LD_PRELOAD=$(clang -print-file-name=libclang_rt.asan-x86_64.so).
_ASAN = os.environ.get("LFMQ_ASAN", "0") == "1"
_CXX_FLAGS = ["-O3", "-std=c++17"] + (
    ["-fsanitize=address", "-fno-omit-frame-pointer"] if _ASAN else [])

SOURCES = [
    "csrc/ext.hip",
    "csrc/pack.hip",
    "csrc/rmsnorm.hip",
    "csrc/rope.hip",
    "csrc/elementwise.hip",
    "csrc/dq_gemv.hip",
    "csrc/dq_gemv_mfma.hip",
    "csrc/dq_gemm.hip",
    "csrc/hessian.hip",
    "csrc/attn_decode.hip",
    "csrc/attn_prefill.hip",
    "csrc/qkv_rope_append.hip",
    "csrc/sample.hip",
    "csrc/probe.hip",
    "csrc/grouped_dq_gemm.hip",
    "csrc/engine.hip",
    "csrc/moe_route.hip",
]

setup(
    name="lfmq_hip",
    ext_modules=[
        CUDAExtension(
            name="lfmq_hip",
            sources=SOURCES,
            extra_compile_args={
                "cxx": _CXX_FLAGS,
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
