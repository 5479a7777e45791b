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
]

ext_modules = [
    cpp_extension.CUDAExtension(
        name="tensor2robot_amd.ops._t2r_hip",
        sources=hip_sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
        },
    ),
]

setup(
    name="tensor2robot_amd",
    version="0.1.0",
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension},
    packages=[],
)
