"""Reference-parity timing harness — mirrors
``regression/benchmark/PerformanceBenchmark.scala``: synthetic
y = sin(sum x / 1000) on 3 features, RBF(0.1), sigma2=1e-3, seed=13,
expert size = active-set size = CLI arg; prints ``TIME: <ms>`` for one fit.

(The repo-level ``bench.py`` is the flagship multi-GPU benchmark; this
script reproduces the reference harness shape exactly.)
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse
import time

from spark_gp_amd import GaussianProcessRegression, RBFKernel
from spark_gp_amd.data import performance_benchmark_data


def main():
    p = argparse.ArgumentParser()
    p.add_argument("expert_size", type=int, nargs="?", default=100)
    p.add_argument("sample_size", type=int, nargs="?", default=100000)
    args = p.parse_args()

    X, y = performance_benchmark_data(args.sample_size, d=3, seed=13)
    gp = (GaussianProcessRegression()
          .setKernel(lambda: RBFKernel(0.1))
          .setDatasetSizeForExpert(args.expert_size)
          .setActiveSetSize(args.expert_size)
          .setSeed(13)
          .setSigma2(1e-3))
    t0 = time.perf_counter()
    gp.fit(X, y)
    ms = (time.perf_counter() - t0) * 1000.0
    print(f"TIME: {ms:.0f}")
    return ms


if __name__ == "__main__":
    main()
