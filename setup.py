"""Build the bagua_amd native extension in-tree for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROCM_HOME = os.environ.get("ROCM_HOME", "/opt/rocm")

ext = cpp_extension.CUDAExtension(
    name="bagua_amd._C",
    sources=[
        "bagua_amd/ops/csrc/core.cpp",
        "bagua_amd/ops/csrc/kernels.hip",
    ],
    include_dirs=[os.path.join(ROCM_HOME, "include")],
    libraries=["rccl"],
    library_dirs=[os.path.join(ROCM_HOME, "lib")],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="bagua_amd",
    version="0.1.0",
    packages=find_packages(include=["bagua_amd", "bagua_amd.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
    entry_points={
        "console_scripts": [
            "baguarun = bagua_amd.distributed.baguarun:main",
            "bagua_sys_perf = bagua_amd.distributed.sys_perf:main",
        ],
    },
)
