"""Build the in-tree gfx950 HIP kernel extension.

Usage (also invoked by __graft_entry__.build()):
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles on GPU-less hosts; the resulting
``pipeedge_amd/ops/_hip_ops*.so`` travels with the repo snapshot.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("pipeedge_amd", "ops", "hip")

ext = CUDAExtension(
    name="pipeedge_amd.ops._hip_ops",
    sources=[
        os.path.join(HIP_DIR, "bindings.cpp"),
        os.path.join(HIP_DIR, "gemm.hip"),
        os.path.join(HIP_DIR, "layernorm.hip"),
        os.path.join(HIP_DIR, "attention.hip"),
        os.path.join(HIP_DIR, "patchify.hip"),
        os.path.join(HIP_DIR, "quantpack.hip"),
        os.path.join(HIP_DIR, "debug_probe.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="pipeedge-amd",
    version="0.1.0",
    packages=["pipeedge_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
