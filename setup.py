"""Build the spark_gp_amd HIP/CDNA4 extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles gfx950 without a GPU present; the resulting
``spark_gp_amd/_hip_ext*.so`` travels with the source tree.
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="spark_gp_amd._hip_ext",
    sources=[
        "spark_gp_amd/ops/csrc/bindings.cpp",
        "spark_gp_amd/ops/csrc/expert_nll.hip",
        "spark_gp_amd/ops/csrc/cross_syrk.hip",
        "spark_gp_amd/ops/csrc/laplace.hip",
        "spark_gp_amd/ops/csrc/big_chol.hip",
        "spark_gp_amd/ops/csrc/synth.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "--offload-arch=gfx950"],
    },
)

setup(
    name="spark_gp_amd",
    version="0.2.0",
    packages=find_packages(include=["spark_gp_amd", "spark_gp_amd.*"]),
    package_data={"spark_gp_amd": ["data/*.csv", "ops/csrc/*"]},
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
