"""GPC benchmark: 1M x 16 ARD binary classification fit (m=1000) on one
MI355X — the config behind profiles/gpc1m_kernel_stats_r01.csv (0.362
s/fit at end of round 1 with the torch fp64 evidence pass).

Round 2 moves the Algorithm 5.1 evidence into the fused Laplace kernel
(K11); this script times fits at a given tol and prints the breakdown.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from spark_gp_amd import GaussianProcessClassifier
from spark_gp_amd.kernels import ARDRBFKernel


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=1_000_000)
    p.add_argument("--dim", type=int, default=16)
    p.add_argument("--active-set", type=int, default=1000)
    p.add_argument("--expert-size", type=int, default=100)
    p.add_argument("--max-iter", type=int, default=15)
    p.add_argument("--tol", type=float, default=1e-5)
    p.add_argument("--fits", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--ppa-precision", type=str, default="fp64",
                   choices=["fp64", "mixed"])
    args = p.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    rng = np.random.default_rng(13)
    X = rng.random((args.rows, args.dim), dtype=np.float64)
    # learnable decision boundary: only 2 active dims at a lengthscale the
    # ARD-RBF can represent with m=1000 (a sin(4*sum of ALL dims) target is
    # label noise at d=16 and lets Newton exit immediately — it would
    # understate the Laplace work)
    y = (np.sin(4.0 * (X[:, 0] + X[:, 1])) > 0).astype(np.float64)
    Xt = torch.tensor(X, dtype=torch.float32, device=dev)
    yt = torch.tensor(y, dtype=torch.float32, device=dev)

    def make():
        return (GaussianProcessClassifier()
                .setKernel(lambda: 1 * ARDRBFKernel(args.dim))
                .setDatasetSizeForExpert(args.expert_size)
                .setActiveSetSize(args.active_set)
                .setSigma2(1e-3)
                .setPpaPrecision(args.ppa_precision)
                .setTol(args.tol)
                .setMaxIter(args.max_iter)
                .setSeed(13)
                .setDevice(dev))

    model = None
    for _ in range(args.warmup):
        model = make().fit(Xt, yt)
    if dev == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.fits):
        model = make().fit(Xt, yt)
    if dev == "cuda":
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.fits
    acc = float((model.predict(Xt[:100_000]) == y[:100_000]).mean())
    print(f"GPC fit: {dt * 1000:.1f} ms/fit  rows={args.rows} "
          f"d={args.dim} m={args.active_set} tol={args.tol} "
          f"maxIter={args.max_iter}  train-acc(100k)={acc:.4f}")
    if model._instr is not None:
        print("stages:", {k: round(v, 4)
                          for k, v in model._instr.timings.items()})


if __name__ == "__main__":
    main()
