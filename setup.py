"""In-tree build of the MI355X-native extension (gfx950 only).

Build: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting easyparallellibrary_amd/_C*.so travels with the repo snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROCM = os.environ.get("ROCM_HOME", "/opt/rocm")

ext = CUDAExtension(
    name="easyparallellibrary_amd._C",
    sources=[
        "csrc/bindings.hip",
        "csrc/comm/rccl_comm.hip",
        "csrc/kernels/kernels.hip",
        "csrc/kernels/moe.hip",
        "csrc/kernels/mfma_probe.hip",
        "csrc/kernels/attention.hip",
    ],
    include_dirs=[os.path.join(ROCM, "include")],
    library_dirs=[os.path.join(ROCM, "lib")],
    libraries=["rccl"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="easyparallellibrary_amd",
    version="0.1.0",
    packages=[
        "easyparallellibrary_amd",
        "easyparallellibrary_amd.strategies",
        "easyparallellibrary_amd.ir",
        "easyparallellibrary_amd.parallel",
        "easyparallellibrary_amd.comm",
        "easyparallellibrary_amd.runtime",
        "easyparallellibrary_amd.ops",
        "easyparallellibrary_amd.models",
        "easyparallellibrary_amd.profiler",
        "easyparallellibrary_amd.utils",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
