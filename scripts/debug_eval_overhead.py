import sys, os, time
sys.path.insert(0, "/root/repo")
import numpy as np, torch, subprocess
subprocess.run(["rocm-smi", "--setperfdeterminism", "2100"], capture_output=True)
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar, compile_kernel
from spark_gp_amd.ops import hip_backend
from spark_gp_amd.models.base import group_experts
from spark_gp_amd.data import shard_performance_benchmark_data
from spark_gp_amd import _hip_ext as ext

X, y = shard_performance_benchmark_data(10_000_000, 32, 0, 1, seed=13)
Xt = torch.as_tensor(X, device="cuda"); yt = torch.as_tensor(y, device="cuda")
t0 = time.perf_counter(); groups = group_experts(Xt, yt, 100); torch.cuda.synchronize()
print("group_experts:", round(time.perf_counter()-t0, 3), "s; groups:",
      [(g[1].shape) for g in groups])
cs = compile_kernel(1 * ARDRBFKernel(32) + Scalar(1e-3).const * EyeKernel())
theta = np.concatenate([[1.0], np.ones(32)])

# warm
for _, Xg, yg in groups:
    hip_backend.nll_grad_compiled(cs, theta, Xg, yg)
torch.cuda.synchronize()

# raw kernel time
_, Xg, yg = groups[0]
sc = torch.ones(32, device="cuda")
t0 = time.perf_counter()
for _ in range(5):
    ext.fused_expert_nll(Xg, yg.float(), sc, 1.0, 1e-3)
torch.cuda.synchronize()
print("raw kernel:", round((time.perf_counter()-t0)/5*1e3, 2), "ms")

# full eval
t0 = time.perf_counter()
for _ in range(5):
    for _, Xg, yg in groups:
        hip_backend.nll_grad_compiled(cs, theta, Xg, yg)
print("full eval:", round((time.perf_counter()-t0)/5*1e3, 2), "ms")

# pieces
import cProfile, pstats, io
pr = cProfile.Profile(); pr.enable()
for _ in range(5):
    for _, Xg, yg in groups:
        hip_backend.nll_grad_compiled(cs, theta, Xg, yg)
pr.disable()
s = io.StringIO(); pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(12)
print("\n".join(s.getvalue().splitlines()[:20]))
