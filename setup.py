"""Build the MI355X HIP extension in-tree.

    python setup.py build_ext --inplace

Cross-compiles for gfx950 via hipcc (no GPU needed to build). The resulting
_smxgb_hip.so lives inside the package so it travels with the source tree.
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", "8")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="sagemaker_xgboost_container_amd",
    version="0.1.0",
    description="MI355X-native SageMaker XGBoost training/serving framework",
    packages=find_packages(exclude=("tests",)),
    ext_modules=[
        CUDAExtension(
            name="sagemaker_xgboost_container_amd.ops._smxgb_hip",
            sources=[
                "sagemaker_xgboost_container_amd/ops/csrc/smxgb_kernels.hip",
                "sagemaker_xgboost_container_amd/ops/csrc/text_parsers.cpp",
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
    entry_points={
        "console_scripts": [
            "serve = sagemaker_xgboost_container_amd.serving:serving_entrypoint",
        ]
    },
)
