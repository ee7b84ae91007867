from .base import (ConstantTimesKernel, EyeKernel, Kernel, Scalar,
                   ScalarTimesKernel, SumOfKernels,
                   TrainableScalarTimesKernel,
                   TrainingVectorsNotInitializedError, WhiteNoiseKernel,
                   sqdist)
from .compiled import CompiledKernel, compile_kernel
from .matern import Matern32Kernel, Matern52Kernel
from .rbf import ARDRBFKernel, RBFKernel

__all__ = [
    "Kernel", "EyeKernel", "WhiteNoiseKernel", "SumOfKernels",
    "ScalarTimesKernel", "ConstantTimesKernel", "TrainableScalarTimesKernel",
    "Scalar", "RBFKernel", "ARDRBFKernel", "Matern32Kernel",
    "Matern52Kernel", "sqdist",
    "TrainingVectorsNotInitializedError", "CompiledKernel", "compile_kernel",
]
