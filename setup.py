"""Build the in-tree HIP/CDNA4 extension for gfx950 (MI355X).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(the arch default is set below; hipcc cross-compiles without a GPU).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join("simple_tip_amd", "ops", "hip")

setup(
    name="simple_tip_amd",
    version="0.1.0",
    packages=[],  # extension-only build; the package itself is used in-tree
    ext_modules=[
        CUDAExtension(
            name="simple_tip_amd.ops._tip_hip",
            sources=[
                os.path.join(HIP_DIR, "bindings.cpp"),
                os.path.join(HIP_DIR, "pairwise.hip"),
                os.path.join(HIP_DIR, "coverage.hip"),
                os.path.join(HIP_DIR, "cam.hip"),
                os.path.join(HIP_DIR, "resnet_fused.hip"),
                os.path.join(HIP_DIR, "scores.hip"),
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
