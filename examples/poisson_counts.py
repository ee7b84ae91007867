"""Count-data GP regression with the Poisson log-link (additive model
family — no reference analog): recover a spatially varying event rate
from Poisson-distributed counts."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from spark_gp_amd import GaussianProcessPoissonRegression
from spark_gp_amd.kernels import ARDRBFKernel


def main():
    rng = np.random.default_rng(0)
    n = 5000
    X = rng.uniform(size=(n, 2))
    true_log_rate = 1.5 + np.sin(3 * X.sum(-1))
    y = rng.poisson(np.exp(true_log_rate)).astype(np.float64)

    model = (GaussianProcessPoissonRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(100)
             .setActiveSetSize(200)
             .setSigma2(1e-2)
             .setMaxIter(30)
             .setSeed(0)
             .fit(X, y))

    rate = model.predict(X[:1000])
    rel = np.abs(rate - np.exp(true_log_rate[:1000])) / np.exp(true_log_rate[:1000])
    print("median relative rate error:", float(np.median(rel)))
    return float(np.median(rel))


if __name__ == "__main__":
    main()
