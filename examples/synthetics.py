"""Synthetic sin-wave regression — mirrors
``regression/examples/Synthetics.scala``: RBF + trainable white noise,
KMeans active set, 10-fold CV, asserts RMSE < 0.11."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from spark_gp_amd import (GaussianProcessRegression, KMeansActiveSetProvider,
                          RBFKernel, WhiteNoiseKernel, cross_validate)
from spark_gp_amd.data import sin_wave


def main():
    X, y = sin_wave(2000, noise_var=0.01, seed=13)

    def gp():
        return (GaussianProcessRegression()
                .setKernel(lambda: 1 * RBFKernel(0.1, 1e-6, 10)
                           + WhiteNoiseKernel(0.5, 0, 1))
                .setDatasetSizeForExpert(100)
                .setActiveSetProvider(KMeansActiveSetProvider())
                .setActiveSetSize(100)
                .setSeed(13)
                .setSigma2(1e-3))

    rmse = cross_validate(gp, X, y, num_folds=10, seed=13)
    print("RMSE:", rmse)
    assert rmse < 0.11
    return rmse


if __name__ == "__main__":
    main()
