"""Airfoil self-noise regression — mirrors
``regression/examples/Airfoil.scala``: ARD-RBF(5) + const noise, expert=100,
active=1000, sigma2=1e-4, 10-fold CV, asserts RMSE < 2.1."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from spark_gp_amd import (GaussianProcessRegression, Scalar, StandardScaler,
                          cross_validate)
from spark_gp_amd.data import load_airfoil
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel


def main():
    X, y = load_airfoil()
    Xs = StandardScaler().fit_transform(X)

    def gp():
        return (GaussianProcessRegression()
                .setDatasetSizeForExpert(100)
                .setActiveSetSize(1000)
                .setSigma2(1e-4)
                .setKernel(lambda: 1 * ARDRBFKernel(5)
                           + Scalar(1).const * EyeKernel()))

    rmse = cross_validate(gp, Xs, y, num_folds=10, seed=13)
    print("RMSE:", rmse)
    assert rmse < 2.1
    return rmse


if __name__ == "__main__":
    main()
