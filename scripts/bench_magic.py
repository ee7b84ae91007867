"""A/B the m x m magic solves: hand-written K13 (big_chol.hip) vs the
torch/rocSOLVER path, same inputs, m=1000 and m=8192."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
import torch

import spark_gp_amd.ppa as ppa
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar

for m in (1000, 8192):
    d = 8
    kernel = 1 * ARDRBFKernel(d) + Scalar(1e-3).const * EyeKernel()
    kernel.set_hyperparameters(np.array([1.1] + [0.9] * d))
    g = torch.Generator(device="cuda").manual_seed(0)
    active = torch.rand(m, d, generator=g, device="cuda")
    B = torch.randn(m, 64, generator=g, dtype=torch.float64, device="cuda")
    KK = B @ B.T + m * torch.eye(m, dtype=torch.float64, device="cuda")
    Ky = torch.randn(m, generator=g, dtype=torch.float64, device="cuda")
    for name, force in (("K13", "0"), ("rocSOLVER", "1")):
        os.environ["SPARK_GP_AMD_FORCE_TORCH"] = force
        mv, mm = ppa.magic_vector_matrix(kernel, KK, Ky, active)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 5 if m <= 1000 else 3
        for _ in range(reps):
            mv, mm = ppa.magic_vector_matrix(kernel, KK, Ky, active)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / reps
        print(f"m={m:5d} {name:>9}: {dt * 1e3:8.2f} ms per magic phase")
    os.environ["SPARK_GP_AMD_FORCE_TORCH"] = "0"
