"""Build the persia_amd HIP extension in-tree (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402
# On ROCm, CUDAExtension drives hipcc; our sources are native HIP (.hip).

this_dir = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="persia_amd._C",
    sources=[
        "persia_amd/csrc/kernels.hip",
        "persia_amd/csrc/dense.hip",
        "persia_amd/csrc/interact.hip",
        "persia_amd/csrc/engine.cpp",
    ],
    include_dirs=[os.path.join(this_dir, "persia_amd", "csrc")],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="persia_amd",
    version="0.1.0",
    packages=find_packages(include=["persia_amd", "persia_amd.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
