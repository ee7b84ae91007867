"""Diagnose the GPU fit quality: A/B hip vs torch objective along the
optimizer path, and the bad-flag fixture."""
import logging
import os
import sys

import numpy as np
import torch

logging.basicConfig(level=logging.INFO)
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from spark_gp_amd import GaussianProcessRegression, rmse
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar, compile_kernel
from spark_gp_amd.ops import hip_backend, torch_backend
from spark_gp_amd.models.base import group_experts

rng = np.random.default_rng(0)
X = rng.random((30000, 8)).astype(np.float32)
y = np.sin(3.0 * X.sum(-1)).astype(np.float32)
Xg = torch.tensor(X, device="cuda")
yg = torch.tensor(y, device="cuda")

cs = compile_kernel(1 * ARDRBFKernel(8) + Scalar(1e-3).const * EyeKernel())
groups = group_experts(Xg, yg, 100)
print("groups:", [(g[1].shape, g[2].shape) for g in groups])

for name, theta in [("x0", np.concatenate([[1.0], np.ones(8)])),
                    ("mid", np.concatenate([[2.0], 0.5 * np.ones(8)]))]:
    for label, fn, Xa, ya in [
        ("hip   ", hip_backend.nll_grad_compiled, Xg, yg),
        ("torch32", torch_backend.nll_grad_compiled, Xg, yg),
        ("oracle ", torch_backend.nll_grad_compiled,
         Xg.double().cpu(), yg.double().cpu()),
    ]:
        tot, grad = 0.0, np.zeros(9)
        for _, Xgg, ygg in group_experts(Xa, ya, 100):
            n, g = fn(cs, theta, Xgg, ygg)
            tot += n
            grad += g
        print(f"{name} {label}: nll={tot:.6e} grad[:4]={grad[:4]}")

# bad-flag fixture
from spark_gp_amd import _hip_ext as ext
g = torch.Generator().manual_seed(9)
Xb = torch.rand(4, 32, 4, generator=g).to("cuda")
Xb[1] = 0.25
Xb[3] = 0.5
yb = torch.sin(Xb.sum(-1))
nll, sumW0, trG, contr, bad = ext.fused_expert_nll(
    Xb, yb, torch.ones(4, device="cuda"), 1.0, 1e-7)
print("bad flags:", bad.cpu().numpy(), "nll:", nll.cpu().numpy())
cs4 = compile_kernel(1 * ARDRBFKernel(4) + Scalar(1e-7).const * EyeKernel())
theta4 = np.concatenate([[1.0], np.ones(4)])
for e in range(4):
    try:
        nll_o, _ = torch_backend.nll_grad_compiled(
            cs4, theta4, Xb[e:e+1].double().cpu(), yb[e:e+1].double().cpu())
    except Exception as exc:
        nll_o = f"exc {exc}"
    print(f"expert {e}: hip={float(nll[e]):.6e} oracle={nll_o}")

# full fits A/B
for force in ["0", "1"]:
    os.environ["SPARK_GP_AMD_FORCE_TORCH"] = force
    gp = (GaussianProcessRegression()
          .setKernel(lambda: 1 * ARDRBFKernel(8))
          .setDatasetSizeForExpert(100).setActiveSetSize(300)
          .setSigma2(1e-3).setMaxIter(15).setSeed(3).setDevice("cuda:0"))
    m = gp.fit(X, y)
    print(f"force_torch={force} rmse={rmse(y[:3000], m.predict(X[:3000])):.4f}",
          "kernel:", repr(m.raw.kernel))
