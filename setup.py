"""Build the cuvite_amd HIP extension in-tree for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="cuvite_amd",
    version="0.1.0",
    packages=["cuvite_amd", "cuvite_amd.ops", "cuvite_amd.parallel",
              "cuvite_amd.utils"],
    ext_modules=[
        CUDAExtension(
            name="cuvite_amd.ops._hip_ops",
            sources=[
                "cuvite_amd/ops/csrc/bindings.cpp",
                "cuvite_amd/ops/csrc/louvain_kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
