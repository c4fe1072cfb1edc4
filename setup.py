"""Build the MI355X-native shuffling data loader's C++/HIP extensions in-tree.

Extensions:
  * ``ray_shuffling_data_loader_amd._rsdl_cpp`` — pure C++ (pybind11) batch
    queue core. Buildable anywhere (no GPU, no torch headers).
  * ``ray_shuffling_data_loader_amd._rsdl_hip`` — HIP/CDNA4 (gfx950) kernels
    for the shuffle hot path, built via torch.utils.cpp_extension (hipcc).
    Cross-compiles on a GPU-less box with PYTORCH_ROCM_ARCH=gfx950.

Usage:  python setup.py build_ext --inplace
"""

import os


from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11
from setuptools import Extension

ext_modules = [
    Extension(
        "ray_shuffling_data_loader_amd._rsdl_cpp",
        sources=["csrc/batch_queue.cpp"],
        include_dirs=[pybind11.get_include()],
        language="c++",
        extra_compile_args=["-O3", "-std=c++17", "-fvisibility=hidden"],
    )
]

cmdclass = {}

# The HIP extension needs torch's extension machinery (drives hipcc for .hip
# sources). Gate on availability so the queue core stays buildable alone.
try:
    from torch.utils import cpp_extension as torch_cpp_ext

    hip_sources = [
        "csrc/shuffle_ops.cpp",
        "csrc/shuffle_kernels.hip",
        "csrc/wgrad_kernel.hip",
        "csrc/relu_bwd.hip",
        "csrc/fwd_chain.hip",
        "csrc/bwd_chain.hip",
        "csrc/wgrad_wide.hip",
        "csrc/wgrad_frag.hip",
    ]
    if all(os.path.exists(s) for s in hip_sources):
        ext_modules.append(
            torch_cpp_ext.CUDAExtension(
                name="ray_shuffling_data_loader_amd._rsdl_hip",
                sources=hip_sources,
                extra_compile_args={
                    "cxx": ["-O3", "-std=c++17"],
                    "nvcc": ["-O3", "-std=c++17"],
                },
            )
        )
        cmdclass["build_ext"] = torch_cpp_ext.BuildExtension
except ImportError:
    pass

setup(
    name="ray_shuffling_data_loader_amd",
    version="0.1.0",
    description=(
        "MI355X-native per-epoch shuffling data loader "
        "(HIP/CDNA4 + RCCL over xGMI)"
    ),
    packages=[
        "ray_shuffling_data_loader_amd",
        "ray_shuffling_data_loader_amd.models",
        "ray_shuffling_data_loader_amd.ops",
        "ray_shuffling_data_loader_amd.parallel",
        "ray_shuffling_data_loader_amd.utils",
    ],
    ext_modules=ext_modules,
    cmdclass=cmdclass,
)
