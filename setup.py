"""In-tree build of the native extensions (HIP kernels for gfx950).

Build:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands inside tensor2robot_amd/ops/ and travels with the repo
snapshot to GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "tensor2robot_amd", "ops", "hip")

hip_sources = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "fused_bn_relu.hip"),
    os.path.join(HIP_DIR, "preprocess.hip"),
    os.path.join(HIP_DIR, "maxpool.hip"),
    os.path.join(HIP_DIR, "spatial_softmax.hip"),
    os.path.join(HIP_DIR, "mfma_probe.hip"),
    os.path.join(HIP_DIR, "conv_s1.hip"),
    os.path.join(HIP_DIR, "conv_s1_big.hip"),
    os.path.join(HIP_DIR, "mdn_nll.hip"),
    os.path.join(HIP_DIR, "conv_wrw.hip"),
    os.path.join(HIP_DIR, "conv_wrw2.hip"),
    os.path.join(HIP_DIR, "conv_wrw4.hip"),
    os.path.join(HIP_DIR, "im2col.hip"),
    os.path.join(HIP_DIR, "jpeg_gpu.hip"),
    os.path.join(HIP_DIR, "conv_stem.hip"),
]

import pybind11
from setuptools import Extension

NATIVE_DIR = os.path.join(ROOT, "tensor2robot_amd", "data", "native")

ext_modules = [
    cpp_extension.CUDAExtension(
        name="tensor2robot_amd.ops._t2r_hip",
        sources=hip_sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
        },
    ),
    # CPU-only data codecs (baseline JPEG): plain pybind11, no torch dep.
    Extension(
        name="tensor2robot_amd.ops._t2r_native",
        sources=[os.path.join(NATIVE_DIR, "jpeg_codec.cpp"),
                 os.path.join(NATIVE_DIR, "example_codec.cpp"),
                 os.path.join(NATIVE_DIR, "native_bindings.cpp")],
        include_dirs=[pybind11.get_include()],
        extra_compile_args=["-O3", "-std=c++17", "-msse4.2"],
        language="c++",
    ),
]

setup(
    name="tensor2robot_amd",
    version="0.1.0",
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension},
    packages=[],
)
