"""Build the native extensions in-tree:

    python setup.py build_ext --inplace

- ``lakesoul_amd._cpp``  — host core (parquet IO, murmur3, CPU decode)
- ``lakesoul_amd._hip``  — gfx950 HIP kernels (decode, merge, hash, ANN);
  built whenever a HIP compiler is present (cross-compiles fine on a
  CPU-only box with PYTORCH_ROCM_ARCH=gfx950).
"""

import os
import sys

from setuptools import setup

import torch
from torch.utils.cpp_extension import BuildExtension, CppExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", str(min(8, os.cpu_count() or 8)))

ROOT = os.path.dirname(os.path.abspath(__file__))

ext_modules = [
    CppExtension(
        "lakesoul_amd._cpp",
        [os.path.join("csrc", "cpp", "module.cc")],
        extra_compile_args=["-O3", "-std=c++17"],
        extra_link_args=["-l:libzstd.so.1"],
    )
]

_WITH_HIP = torch.version.hip is not None and os.path.exists(
    os.path.join(ROOT, "csrc", "hip", "hip_module.cc")
)
if _WITH_HIP:
    from torch.utils.cpp_extension import CUDAExtension  # maps to hip on ROCm

    ext_modules.append(
        CUDAExtension(
            "lakesoul_amd._hip",
            [
                os.path.join("csrc", "hip", "hip_module.cc"),
                os.path.join("csrc", "hip", "kernels.hip"),
                os.path.join("csrc", "hip", "ann.hip"),
                os.path.join("csrc", "hip", "fastscan.hip"),
                os.path.join("csrc", "hip", "snappy.hip"),
                os.path.join("csrc", "hip", "zstd.hip"),
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    )

setup(
    name="lakesoul_amd",
    version="0.1.0",
    packages=[
        "lakesoul_amd",
        "lakesoul_amd.meta",
        "lakesoul_amd.io",
        "lakesoul_amd.tables",
        "lakesoul_amd.parallel",
        "lakesoul_amd.torch",
        "lakesoul_amd.arrow",
        "lakesoul_amd.vector",
        "lakesoul_amd.ops",
        "lakesoul_amd.utils",
        "lakesoul_amd.service",
    ],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
