"""Per-phase timing of the fused expert NLL kernel (wall_clock64, 100 MHz)."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from spark_gp_amd import _hip_ext as ext

E, k, d = 20000, 100, 32
g = torch.Generator().manual_seed(0)
X = torch.rand(E, k, d, generator=g).cuda()
y = torch.sin(3 * X.sum(-1)).cuda()
scale = torch.rand(d, generator=g).add(0.5).cuda()

# warmup + DVFS clock ramp (sustained load for ~10s)
t0 = time.perf_counter()
while time.perf_counter() - t0 < 10.0:
    out = ext.fused_expert_nll(X, y, scale, 1.0, 1e-3)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(5):
    out = ext.fused_expert_nll(X, y, scale, 1.0, 1e-3)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 5
print(f"E={E} k={k} d={d}: {dt*1e3:.2f} ms/launch, "
      f"{dt/E*1e6:.2f} us/expert-slot")

*_, clk = ext.fused_expert_nll_profile(X, y, scale, 1.0, 1e-3, True)
clk = clk.cpu().numpy().astype(np.float64)
# deltas[i] = clk[i+1]-clk[i] = duration of phase phases[i]
phases = ["A stage", "B build", "C chol", "D trtri", "E alpha", "L lauum",
          "W w0", "G rowsum", "H contr"]
names = phases
# per-expert deltas in microseconds (wall_clock64 = 100 MHz)
FREQ = 1e8
deltas = (clk[:, 1:10] - clk[:, 0:9]) / FREQ * 1e6
mean = deltas.mean(0)
total = mean.sum()
print(f"sum of phases: {total:.2f} us (mean per expert)")
for i in range(9):
    print(f"  {phases[i]:>8}: {mean[i]:8.2f} us  ({100*mean[i]/total:4.1f}%)")
sub = clk[:, 12:18].mean(0) / FREQ * 1e6
subnames = ["phase1 diag-pre", "phase2 wall", "wave0 span",
            "waves1-7 span", "C2 panel solve", "D off-diag trtri"]
print("C/D sub-phases (accumulated over J, mean per expert):")
for n, v in zip(subnames, sub):
    print(f"  {n:>18}: {v:8.2f} us")
# wall span of the whole launch from clocks
span = (clk[:, 9].max() - clk[:, 0].min()) / FREQ * 1e3
print(f"launch span by clocks: {span:.2f} ms")

# contention probe: few blocks (1 per CU) vs full load
for Ee in (256, 2048):
    Xs, ys = X[:Ee], y[:Ee]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        ext.fused_expert_nll(Xs, ys, scale, 1.0, 1e-3)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    *_, clk2 = ext.fused_expert_nll_profile(Xs, ys, scale, 1.0, 1e-3, True)
    clk2 = clk2.cpu().numpy().astype(np.float64)
    dd = (clk2[:, 1:10] - clk2[:, 0:9]) / 1e8 * 1e6
    m = dd.mean(0)
    print(f"E={Ee}: {dt*1e3:.3f} ms/launch; phases us: "
          + " ".join(f"{phases[i]}:{m[i]:.1f}" for i in range(9)))
