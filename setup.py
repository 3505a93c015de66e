"""In-tree build of the native components.

Two extensions, built with `python setup.py build_ext --inplace` so the .so
files live inside the package and travel with the repo snapshot:
  * llm_d_inference_scheduler_amd._router_core  — C++ router core (pybind11,
    CPU): prefix index + chained hashing, scheduler hot loop, flow queues.
  * llm_d_inference_scheduler_amd._hip_ops      — gfx950 HIP kernels (torch
    extension, hipcc --offload-arch=gfx950): prefix hash/match, paged KV,
    fused norm/rope/act, GQA decode attention.
"""
import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from pybind11.setup_helpers import Pybind11Extension
from torch.utils import cpp_extension

ROOT = os.path.dirname(os.path.abspath(__file__))

router_core = Pybind11Extension(
    "llm_d_inference_scheduler_amd._router_core",
    sources=["csrc/router/bindings.cpp"],
    cxx_std=17,
    extra_compile_args=["-O3"],
)

hip_ops = cpp_extension.CUDAExtension(
    name="llm_d_inference_scheduler_amd._hip_ops",
    sources=[
        "csrc/hip/ops.hip",
        "csrc/hip/prefix_kernels.hip",
        "csrc/hip/norm_rope_act.hip",
        "csrc/hip/kv_cache.hip",
        "csrc/hip/paged_attention.hip",
        "csrc/hip/flash_prefill.hip",
        "csrc/hip/flash_prefill_glds.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
    },
)

ext_modules = [router_core]
if os.environ.get("LDS_AMD_SKIP_HIP", "0") != "1":
    ext_modules.append(hip_ops)

setup(
    name="llm_d_inference_scheduler_amd",
    version="0.1.0",
    packages=["llm_d_inference_scheduler_amd"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(use_ninja=True)},
)
