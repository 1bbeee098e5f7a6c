"""Build the in-tree HIP/gfx950 extension:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces seist_amd/_C.*.so next to the package so the GPU-box snapshot
carries it (no JIT cache dependence).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("seist_amd", "ops", "hip")

# K20: plain-C++ data-pipeline workers (no HIP/torch dependency — runs in
# DataLoader worker processes)
native_data = Extension(
    name="seist_amd._native_data",
    sources=[os.path.join("seist_amd", "data", "_native.cpp"),
             os.path.join("seist_amd", "data", "_augment.cpp")],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-std=c++17"],
    language="c++",
)

ext = CUDAExtension(
    name="seist_amd._C",
    sources=[
        os.path.join(HIP_DIR, "bindings.cpp"),
        os.path.join(HIP_DIR, "pw_conv.hip"),
        os.path.join(HIP_DIR, "conv1d.hip"),
        os.path.join(HIP_DIR, "bn_act.hip"),
        os.path.join(HIP_DIR, "pool_interp.hip"),
        os.path.join(HIP_DIR, "adam.hip"),
        os.path.join(HIP_DIR, "reductions.hip"),
        os.path.join(HIP_DIR, "pw_mfma.hip"),
        os.path.join(HIP_DIR, "dw_mfma.hip"),
        os.path.join(HIP_DIR, "conv_tap.hip"),
        os.path.join(HIP_DIR, "conv_smallc.hip"),
        os.path.join(HIP_DIR, "conv_dw_smallc.hip"),
        os.path.join(HIP_DIR, "rowscale.hip"),
        os.path.join(HIP_DIR, "attention.hip"),
        os.path.join(HIP_DIR, "eqt.hip"),
        os.path.join(HIP_DIR, "loss.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="seist_amd",
    version="0.1.0",
    packages=["seist_amd"],
    ext_modules=[ext, native_data],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
