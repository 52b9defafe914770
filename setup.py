"""Builds the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting marian_amd/ops/_hip_ext*.so is git-ignored but travels with
gpurun snapshots to the GPU box.
"""
import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

HIP_DIR = os.path.join("marian_amd", "ops", "hip")

sources = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "layernorm.hip"),
    os.path.join(HIP_DIR, "softmax.hip"),
    os.path.join(HIP_DIR, "cross_entropy.hip"),
    os.path.join(HIP_DIR, "adam.hip"),
    os.path.join(HIP_DIR, "wgrad.hip"),
]
attn = os.path.join(HIP_DIR, "attention.hip")
if os.path.exists(attn):
    sources.append(attn)
topk = os.path.join(HIP_DIR, "topk.hip")
if os.path.exists(topk):
    sources.append(topk)
colsum = os.path.join(HIP_DIR, "colsum.hip")
if os.path.exists(colsum):
    sources.append(colsum)
fp8q = os.path.join(HIP_DIR, "fp8q.hip")
if os.path.exists(fp8q):
    sources.append(fp8q)
ltgemm = os.path.join(HIP_DIR, "ltgemm.cpp")
if os.path.exists(ltgemm):
    sources.append(ltgemm)

ext = CUDAExtension(
    name="marian_amd.ops._hip_ext",
    sources=sources,
    libraries=["hipblaslt"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="marian_amd",
    version="0.1.0",
    packages=["marian_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
