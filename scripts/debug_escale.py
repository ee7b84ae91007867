import sys, time, subprocess
sys.path.insert(0, "/root/repo")
import torch
subprocess.run(["rocm-smi", "--setperfdeterminism", "2100"], capture_output=True)
from spark_gp_amd import _hip_ext as ext
g = torch.Generator().manual_seed(0)
Xall = torch.rand(100000, 100, 32, generator=g).cuda()
yall = torch.sin(3*Xall.sum(-1)).cuda()
sc = torch.rand(32, generator=g).add(0.5).cuda()
def run(E, n=4):
    X, y = Xall[:E], yall[:E]
    for _ in range(2): ext.fused_expert_nll(X, y, sc, 1.0, 1e-3)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): ext.fused_expert_nll(X, y, sc, 1.0, 1e-3)
    torch.cuda.synchronize()
    dt = (time.perf_counter()-t0)/n
    print(f"E={E:6d}: {dt*1e3:7.2f} ms  {dt/E*512*1e6:6.1f} us/expert-round")
for E in (2000, 5000, 10000, 20000, 40000, 70000, 100000, 20000, 10000):
    run(E)
