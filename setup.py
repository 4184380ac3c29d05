"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

produces flake16_framework_amd/ops/_hip.*.so next to the package sources so
the repo snapshot carries it to the GPU box.  -ffp-contract=off is required:
fp64 split scores and fp32 SMOTE interpolation must be bit-identical to the
numpy reference (models/forest_ref.py, balance/__init__.py).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="flake16_framework_amd_hip",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="flake16_framework_amd.ops._hip",
            sources=["flake16_framework_amd/ops/hip/module.hip"],
            depends=["flake16_framework_amd/ops/hip/forest.hip",
                     "flake16_framework_amd/ops/hip/knn_balance.hip",
                     "flake16_framework_amd/ops/hip/scaler_pca.hip",
                     "flake16_framework_amd/ops/hip/treeshap.hip",
                     "flake16_framework_amd/ops/hip/philox.h"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "-ffp-contract=off"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
