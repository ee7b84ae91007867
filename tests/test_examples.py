"""End-to-end quality gates inherited from the reference's asserting examples
(SURVEY.md §4): Synthetics 10-fold CV RMSE < 0.11
(``regression/examples/Synthetics.scala:33``) and Airfoil 10-fold CV RMSE
< 2.1 (``regression/examples/Airfoil.scala:24``), both on CPU world_size=1
(BASELINE config 1)."""

import numpy as np
import pytest

from spark_gp_amd import (GaussianProcessRegression, KMeansActiveSetProvider,
                          RBFKernel, Scalar, StandardScaler, WhiteNoiseKernel,
                          cross_validate)
from spark_gp_amd.data import load_airfoil, sin_wave
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel


@pytest.mark.slow
def test_synthetics_gate_rmse():
    X, y = sin_wave(2000, noise_var=0.01, seed=13)

    def factory():
        return (GaussianProcessRegression()
                .setKernel(lambda: 1 * RBFKernel(0.1, 1e-6, 10)
                           + WhiteNoiseKernel(0.5, 0, 1))
                .setDatasetSizeForExpert(100)
                .setActiveSetProvider(KMeansActiveSetProvider())
                .setActiveSetSize(100)
                .setSeed(13)
                .setSigma2(1e-3)
                .setDevice("cpu"))

    rmse_cv = cross_validate(factory, X, y, num_folds=10, seed=13)
    print("Synthetics RMSE:", rmse_cv)
    assert rmse_cv < 0.11


@pytest.mark.slow
def test_airfoil_gate_rmse():
    X, y = load_airfoil()
    Xs = StandardScaler().fit_transform(X)
    ys = y  # reference scales features only (labels pass through Scaling
            # with the features; its scale() standardizes features, keeps y)

    def factory():
        return (GaussianProcessRegression()
                .setKernel(lambda: 1 * ARDRBFKernel(5)
                           + Scalar(1).const * EyeKernel())
                .setDatasetSizeForExpert(100)
                .setActiveSetSize(1000)
                .setSigma2(1e-4)
                .setSeed(0)
                .setDevice("cpu"))

    rmse_cv = cross_validate(factory, Xs, ys, num_folds=10, seed=13)
    print("Airfoil RMSE:", rmse_cv)
    assert rmse_cv < 2.1
