import sys; sys.path.insert(0, "/root/repo")
import torch
from spark_gp_amd import _hip_ext as ext
E, k, d = 20000, 100, 32
g = torch.Generator().manual_seed(0)
X = torch.rand(E, k, d, generator=g).cuda()
y = torch.sin(3*X.sum(-1)).cuda()
scale = torch.rand(d, generator=g).add(0.5).cuda()
for _ in range(3):
    ext.fused_expert_nll(X, y, scale, 1.0, 1e-3)
torch.cuda.synchronize()
