import sys, os
sys.path.insert(0, "/root/repo")
import numpy as np, torch
from spark_gp_amd import _hip_ext as ext
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar, compile_kernel
from spark_gp_amd.ops import torch_backend

for k in [32, 64, 96, 100, 128]:
    d = 8
    E = 4
    g = torch.Generator().manual_seed(1)
    X = torch.rand(E, k, d, generator=g).cuda()
    y = torch.sin(3*X.sum(-1)).cuda()
    scale = torch.ones(d).cuda()
    nll, sw, tg, ct, bad = ext.fused_expert_nll(X, y, scale, 1.0, 1e-3)
    cs = compile_kernel(1*ARDRBFKernel(d) + Scalar(1e-3).const*EyeKernel())
    theta = np.concatenate([[1.0], np.ones(d)])
    no, go = torch_backend.nll_grad_compiled(cs, theta, X.double().cpu(), y.double().cpu())
    nh = float(nll.sum())
    print(f"k={k}: bad={bad.cpu().numpy()} nll_hip={nh:.6f} nll_oracle={no:.6f} "
          f"rel={(nh-no)/abs(no):.2e} sw_rel={(float(sw.sum())) :.4f} ")
