"""Build the torchmpi_amd native extension in-tree.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""
import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils import cpp_extension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "torchmpi_amd", "csrc")

sources = [
    os.path.join(CSRC, "module.cpp"),
    os.path.join(CSRC, "tcp.cpp"),
    os.path.join(CSRC, "cpu_comm.cpp"),
    os.path.join(CSRC, "ps.cpp"),
    os.path.join(CSRC, "rccl_engine.hip"),
    os.path.join(CSRC, "ipc_ps.hip"),
    os.path.join(CSRC, "kernels", "reduce.hip"),
    os.path.join(CSRC, "kernels", "pack.hip"),
    os.path.join(CSRC, "kernels", "batchnorm.hip"),
    os.path.join(CSRC, "kernels", "rmsnorm.hip"),
]

extra_cxx = ["-O3", "-std=c++17", "-Wall"]
extra_hip = ["-O3", "-std=c++17"]

ext = cpp_extension.CUDAExtension(
    name="torchmpi_amd._C",
    sources=sources,
    include_dirs=[CSRC, "/opt/rocm/include", "/opt/rocm/include/rccl"],
    libraries=["rccl"],
    library_dirs=["/opt/rocm/lib"],
    extra_compile_args={"cxx": extra_cxx, "nvcc": extra_hip},
)

setup(
    name="torchmpi_amd",
    version="0.1.0",
    packages=["torchmpi_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
