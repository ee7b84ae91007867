"""Matérn 3/2 and 5/2 kernels (additive family; no reference analog):
closed-form checks, finite-difference derivative checks, DSL composition,
end-to-end fit through the generic objective path, and model IO round-trip.
"""

import math

import numpy as np
import pytest
import torch

from spark_gp_amd import GaussianProcessRegression
from spark_gp_amd.kernels import (EyeKernel, Matern32Kernel, Matern52Kernel,
                                  Scalar, compile_kernel)
from tests.test_kernels import fd_gradient_check


@pytest.fixture
def X4():
    rng = np.random.default_rng(3)
    return torch.as_tensor(rng.normal(size=(5, 3)))


def brute_matern(X, Z, l, nu):
    X, Z = X.numpy(), Z.numpy()
    K = np.zeros((len(X), len(Z)))
    for i, a in enumerate(X):
        for j, b in enumerate(Z):
            r = np.linalg.norm(a - b)
            if nu == 1.5:
                s = math.sqrt(3.0) * r / l
                K[i, j] = (1 + s) * math.exp(-s)
            else:
                s = math.sqrt(5.0) * r / l
                K[i, j] = (1 + s + s * s / 3) * math.exp(-s)
    return K


@pytest.mark.parametrize("cls,nu", [(Matern32Kernel, 1.5),
                                    (Matern52Kernel, 2.5)])
def test_matern_matches_bruteforce(X4, cls, nu):
    k = cls(0.8)
    np.testing.assert_allclose(k.training_kernel(X4).numpy(),
                               brute_matern(X4, X4, 0.8, nu), atol=1e-10)
    Z = X4[:2] + 0.3
    np.testing.assert_allclose(k.cross_kernel(Z, X4).numpy(),
                               brute_matern(Z, X4, 0.8, nu), atol=1e-10)
    assert torch.allclose(k.training_kernel_diag(X4),
                          torch.ones(5, dtype=X4.dtype))


@pytest.mark.parametrize("cls", [Matern32Kernel, Matern52Kernel])
def test_matern_derivative_fd(X4, cls):
    fd_gradient_check(cls(0.7), X4)


def test_matern_composes_in_dsl(X4):
    k = (2.0 * Matern32Kernel(1.0) + Matern52Kernel(2.0)
         + Scalar(1e-3).const * EyeKernel())
    assert k.num_hyperparameters == 3          # C (prepended) + 2 lengthscales
    fd_gradient_check(k, X4)
    assert compile_kernel(k) is None           # no fused fast path: generic


def test_matern_fit_end_to_end():
    rng = np.random.default_rng(0)
    X = rng.uniform(size=(400, 2))
    y = np.sin(3 * X.sum(-1)) + 0.05 * rng.normal(size=400)
    m = (GaussianProcessRegression()
         .setKernel(lambda: 1 * Matern52Kernel(1.0))
         .setDatasetSizeForExpert(50).setActiveSetSize(80)
         .setSigma2(1e-2).setMaxIter(30).setSeed(0).setDevice("cpu")
         .fit(X, y))
    from spark_gp_amd.utils import rmse
    assert rmse(y, m.predict(X)) < 0.15


def test_matern_model_io_roundtrip(tmp_path):
    from spark_gp_amd.models.model_io import kernel_from_spec, kernel_to_spec
    for k in (Matern32Kernel(0.5, 1e-3, 10.0), Matern52Kernel(2.5)):
        k2 = kernel_from_spec(kernel_to_spec(k))
        assert type(k2) is type(k) and k2.l == k.l
        assert (k2.lower, k2.upper) == (k.lower, k.upper)
